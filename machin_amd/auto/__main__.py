"""Auto-layer CLI.

Parity target: reference ``machin/auto/__main__.py`` (:13-96):

    python -m machin_amd.auto generate --algo DQN --env CartPole-v1 \
        --print / --output config.json
    python -m machin_amd.auto launch --config config.json
"""
import argparse
import json
import sys

from ..utils.conf import Config, load_config_file, save_config
from .config import (
    generate_algorithm_config,
    generate_env_config,
    generate_training_config,
    get_available_algorithms,
    get_available_environments,
    launch,
)


def main(argv=None):
    parser = argparse.ArgumentParser(prog="python -m machin_amd.auto")
    sub = parser.add_subparsers(dest="command", required=True)

    gen = sub.add_parser("generate", help="generate a launch config")
    gen.add_argument("--algo", required=True,
                     choices=get_available_algorithms())
    gen.add_argument("--env", required=True,
                     choices=get_available_environments())
    gen.add_argument("--output", default=None, help="output JSON path")
    gen.add_argument("--print", dest="print_", action="store_true")

    lau = sub.add_parser("launch", help="launch training from a config")
    lau.add_argument("--config", required=True)

    lst = sub.add_parser("list", help="list algorithms / environments")

    args = parser.parse_args(argv)

    if args.command == "generate":
        config = generate_env_config(args.env)
        config = generate_algorithm_config(args.algo, config)
        config = generate_training_config(config=config)
        if args.output:
            save_config(config, args.output)
            print(f"Config written to {args.output}")
        if args.print_ or not args.output:
            print(json.dumps(config.data, indent=2, default=str))
        return 0

    if args.command == "launch":
        config = load_config_file(args.config)
        launch(config)
        return 0

    if args.command == "list":
        print("algorithms:", ", ".join(get_available_algorithms()))
        print("environments:", ", ".join(get_available_environments()))
        return 0


if __name__ == "__main__":
    sys.exit(main())
