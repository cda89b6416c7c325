"""MADDPG: multi-agent DDPG with centralized critics.

Parity target: reference ``machin/frame/algorithms/maddpg.py``
(:45-1066): per-agent actor/critic pairs with target networks,
ensemble sub-policies (``sub_policy_num``), a ``critic_visible_actors``
observability matrix (:106-113), lockstep per-agent replay with
same-index sampling (:959-966), pool-parallel per-agent updates
(:594-634): ``pool_type="thread"`` runs them on a thread pool
(models share one GPU, no IPC copies); ``pool_type="process"`` runs
them on a :class:`machin_amd.parallel.pool.P2PPool` with every model
parameter AND optimizer state tensor in shared memory, so worker
updates mutate the parent's networks in place (reference's
shared-memory pool mode, maddpg.py:594-634 + SHMBuffer :16-44).
Action/state concat hooks (:968-994).
"""
import copy
import random
from typing import Any, Callable, Dict, List, Union

import numpy as np
import torch as t
import torch.nn as nn

from ...parallel.pool import P2PPool, ThreadPool
from ...utils.conf import Config
from ..buffers.buffer import Buffer
from ..noise.action_space_noise import (
    add_clipped_normal_noise_to_action,
    add_normal_noise_to_action,
    add_ou_noise_to_action,
    add_uniform_noise_to_action,
)
from .base import TorchFramework
from .utils import hard_update, safe_call, safe_return, soft_update


def _share_optimizer_state(opt):
    """Materialize optimizer state with a zero-grad step, then move
    every state tensor into shared memory so P2PPool workers' steps
    accumulate into the PARENT's Adam moments (a worker receives the
    optimizer by reference-serialization each call)."""
    saved_wd = []
    for group in opt.param_groups:
        # weight decay would turn the materialization step into a
        # real parameter update even with zero gradients
        saved_wd.append(group.get("weight_decay", 0))
        group["weight_decay"] = 0
        for p in group["params"]:
            if p.grad is None:
                p.grad = t.zeros_like(p)
    opt.step()  # zero grad + zero wd: params unchanged, state built
    opt.zero_grad(set_to_none=False)
    for group, wd in zip(opt.param_groups, saved_wd):
        group["weight_decay"] = wd
    for st in opt.state.values():
        for v in st.values():
            if t.is_tensor(v):
                v.share_memory_()


def _process_update_task(payload):
    """One agent's critic+actor update, executed in a P2PPool worker.
    All parameters/optimizer state arrive as shared-memory references
    (machin_amd.parallel.pickle copy_tensor=False), so the in-place
    optimizer steps are visible to the parent process."""
    (actor, actor_optim, critic, critic_optim, critic_target,
     target_actors, batches_vis, agent_pos, reward, terminal, bsize,
     discount, grad_max, update_value, update_policy, criterion,
     fns) = payload
    atf, acf, scf, rwf = fns
    states = [b[0] for b in batches_vis]
    actions = [b[1] for b in batches_vis]
    next_states = [b[2] for b in batches_vis]

    with t.no_grad():
        next_acts = [
            atf(safe_return(safe_call(pol, ns)))
            for pol, ns in zip(target_actors, next_states)
        ]
        next_value = safe_return(
            safe_call(critic_target, scf(next_states), acf(next_acts))
        ).view(bsize, 1)
        device = next_value.device
        rew = reward.to(device).float().view(bsize, 1)
        term = terminal.to(device).float().view(bsize, 1)
        y = rwf(rew, discount, next_value, term)

    all_states = scf(states)
    cur_value = safe_return(
        safe_call(
            critic, all_states,
            acf([{"action": a["action"]} for a in actions]),
        )
    ).view(bsize, 1)
    value_loss = criterion(cur_value, y.to(cur_value.dtype))
    if update_value:
        critic_optim.zero_grad(set_to_none=True)
        value_loss.backward()
        nn.utils.clip_grad_norm_(critic.parameters(), grad_max)
        critic_optim.step()

    cur_action = atf(safe_return(safe_call(actor, states[agent_pos])))
    policy_actions = [
        cur_action if i == agent_pos else {"action": actions[i]["action"]}
        for i in range(len(batches_vis))
    ]
    act_value = safe_return(
        safe_call(critic, all_states, acf(policy_actions))
    )
    act_policy_loss = -act_value.mean()
    if update_policy:
        actor_optim.zero_grad(set_to_none=True)
        act_policy_loss.backward()
        nn.utils.clip_grad_norm_(actor.parameters(), grad_max)
        actor_optim.step()

    return (
        -float(act_policy_loss.detach().item()),
        float(value_loss.detach().item()),
    )


class MADDPG(TorchFramework):
    _is_top = ["all_actor_target", "all_critic_target"]
    _is_restorable = ["all_actor_target", "all_critic_target"]

    def __init__(
        self,
        actors: List[nn.Module],
        actor_targets: List[nn.Module],
        critics: List[nn.Module],
        critic_targets: List[nn.Module],
        optimizer: Callable,
        criterion: Callable,
        *_,
        lr_scheduler: Callable = None,
        lr_scheduler_args=None,
        lr_scheduler_kwargs=None,
        critic_visible_actors: List[List[int]] = None,
        sub_policy_num: int = 0,
        batch_size: int = 100,
        update_rate: float = 0.001,
        update_steps: Union[int, None] = None,
        actor_learning_rate: float = 0.0005,
        critic_learning_rate: float = 0.001,
        discount: float = 0.99,
        gradient_max: float = np.inf,
        replay_size: int = 500000,
        replay_device: Union[str, t.device] = "cpu",
        replay_buffer: Buffer = None,
        visualize: bool = False,
        visualize_dir: str = "",
        use_jit: bool = False,
        pool_type: str = "thread",
        pool_size: int = None,
        **__,
    ):
        super().__init__()
        if not (
            len(actors) == len(actor_targets) == len(critics)
            == len(critic_targets)
        ):
            raise ValueError("Actor/critic list lengths must match.")
        n = len(actors)
        self.agent_num = n
        self.batch_size = batch_size
        self.update_rate = update_rate
        self.update_steps = update_steps
        self.discount = discount
        self.grad_max = gradient_max
        self.visualize = visualize
        self.visualize_dir = visualize_dir
        self._update_counter = 0
        self.critic_visible_actors = (
            critic_visible_actors or [list(range(n))] * n
        )

        # ensemble: each agent gets 1 + sub_policy_num policies
        self.ensemble_size = 1 + sub_policy_num
        self.actors = [
            [a] + [copy.deepcopy(a) for _ in range(sub_policy_num)]
            for a in actors
        ]
        self.actor_targets = [
            [at] + [copy.deepcopy(at) for _ in range(sub_policy_num)]
            for at in actor_targets
        ]
        self.critics = list(critics)
        self.critic_targets = list(critic_targets)

        for agent in range(n):
            for p in range(self.ensemble_size):
                hard_update(
                    self.actor_targets[agent][p], self.actors[agent][p]
                )
            hard_update(self.critic_targets[agent], self.critics[agent])

        self.actor_optims = [
            [
                optimizer(pol.parameters(), lr=actor_learning_rate)
                for pol in agent_pols
            ]
            for agent_pols in self.actors
        ]
        self.critic_optims = [
            optimizer(c.parameters(), lr=critic_learning_rate)
            for c in self.critics
        ]

        # one lockstep buffer per agent
        if replay_buffer is not None:
            raise ValueError(
                "MADDPG manages per-agent buffers internally; custom "
                "replay_buffer is not supported."
            )
        self.replay_buffers = [
            Buffer(replay_size, replay_device) for _ in range(n)
        ]

        self.actor_lr_schs = None
        self.critic_lr_schs = None
        if lr_scheduler is not None:
            a_args, c_args = lr_scheduler_args or ([()] * n, [()] * n)
            a_kw, c_kw = lr_scheduler_kwargs or ([{}] * n, [{}] * n)
            self.actor_lr_schs = [
                lr_scheduler(self.actor_optims[i][0], *a_args[i], **a_kw[i])
                for i in range(n)
            ]
            self.critic_lr_schs = [
                lr_scheduler(self.critic_optims[i], *c_args[i], **c_kw[i])
                for i in range(n)
            ]

        self.criterion = (
            criterion() if isinstance(criterion, type) else criterion
        )
        # TorchScript actors: scripted modules SHARE parameters with
        # the originals (updates through the optimizers are visible),
        # and scripted forwards drop the GIL so act() parallelizes
        # across agents on a thread pool.
        self.use_jit = use_jit
        self._jit_actors = None
        self._jit_actor_targets = None
        self._act_pool = None
        if use_jit:
            try:
                self._jit_actors = [
                    [t.jit.script(p) for p in agent_pols]
                    for agent_pols in self.actors
                ]
                self._jit_actor_targets = [
                    [t.jit.script(p) for p in agent_pols]
                    for agent_pols in self.actor_targets
                ]
            except Exception as e:  # noqa: BLE001 - clearer message
                raise ValueError(
                    "use_jit=True requires TorchScript-compatible "
                    f"actor models: {e}"
                ) from e
            self._act_pool = ThreadPool(processes=min(n, 8))
        if pool_type not in ("thread", "process"):
            raise ValueError(
                f"pool_type must be 'thread' or 'process', got "
                f"{pool_type!r}"
            )
        self.pool_type = pool_type
        if pool_type == "process":
            # share everything the workers mutate
            for group in self.actors + self.actor_targets:
                for mod in group:
                    mod.share_memory()
            for mod in self.critics + self.critic_targets:
                mod.share_memory()
            for opts in self.actor_optims:
                for opt in opts:
                    _share_optimizer_state(opt)
            for opt in self.critic_optims:
                _share_optimizer_state(opt)
            self.pool = P2PPool(
                processes=pool_size or min(n, 8), copy_tensor=False
            )
        else:
            self.pool = ThreadPool(
                processes=pool_size or min(n, 8)
            )

        # checkpoint containers
        self.all_actor_target = nn.Module()
        self.all_critic_target = nn.Module()
        for i in range(n):
            for p in range(self.ensemble_size):
                self.all_actor_target.add_module(
                    f"actor_{i}_{p}", self.actor_targets[i][p]
                )
            self.all_critic_target.add_module(
                f"critic_{i}", self.critic_targets[i]
            )

    # ------------------------------------------------------------------
    @property
    def optimizers(self):
        out = []
        for agent_optims in self.actor_optims:
            out.extend(agent_optims)
        out.extend(self.critic_optims)
        return out

    @optimizers.setter
    def optimizers(self, optimizers):
        idx = 0
        for agent in range(self.agent_num):
            for p in range(self.ensemble_size):
                self.actor_optims[agent][p] = optimizers[idx]
                idx += 1
        for agent in range(self.agent_num):
            self.critic_optims[agent] = optimizers[idx]
            idx += 1

    @property
    def lr_schedulers(self):
        out = []
        if self.actor_lr_schs is not None:
            out.extend(self.actor_lr_schs)
        if self.critic_lr_schs is not None:
            out.extend(self.critic_lr_schs)
        return out

    # ------------------------------------------------------------------
    # acting (one state dict per agent)
    # ------------------------------------------------------------------
    def _policies(self, use_target: bool):
        """A randomly drawn ensemble member per agent."""
        src = self.actor_targets if use_target else self.actors
        return [random.choice(agent_pols) for agent_pols in src]

    def _jit_policies(self, use_target: bool):
        src = self._jit_actor_targets if use_target else self._jit_actors
        return [random.choice(agent_pols) for agent_pols in src]

    def act(self, states: List[Dict[str, Any]], use_target: bool = False,
            **__):
        """Returns a list of action tensors, one per agent.

        With ``use_jit`` the per-agent forwards run CONCURRENTLY on a
        thread pool: TorchScript forwards release the GIL, so N agents
        act in parallel on one device (reference
        machin/frame/algorithms/maddpg.py:278-303; the scripted
        modules share parameter memory with the trained actors)."""
        if self.use_jit:
            pols = self._jit_policies(use_target)

            def _one(pol, st):
                with t.no_grad():
                    return safe_return(pol(**st))

            return self._act_pool.starmap(
                _one, list(zip(pols, states))
            )
        pols = self._policies(use_target)
        with t.no_grad():
            return [
                safe_return(safe_call(pol, st))
                for pol, st in zip(pols, states)
            ]

    def act_with_noise(
        self,
        states: List[Dict[str, Any]],
        noise_param: Any = (0.0, 1.0),
        ratio: float = 1.0,
        mode: str = "uniform",
        use_target: bool = False,
        **__,
    ):
        actions = self.act(states, use_target)
        adders = {
            "uniform": add_uniform_noise_to_action,
            "normal": add_normal_noise_to_action,
            "clipped_normal": add_clipped_normal_noise_to_action,
            "ou": add_ou_noise_to_action,
        }
        if mode not in adders:
            raise ValueError(f"Unknown noise type {mode!r}")
        return [adders[mode](a, noise_param, ratio) for a in actions]

    def act_discrete(self, states: List[Dict[str, Any]],
                     use_target: bool = False, **__):
        actions = self.act(states, use_target)
        out_a, out_p = [], []
        for a in actions:
            out_a.append(t.argmax(a, dim=1).view(a.shape[0], 1))
            out_p.append(a)
        return out_a, out_p

    def act_discrete_with_noise(self, states: List[Dict[str, Any]],
                                use_target: bool = False, **__):
        actions = self.act(states, use_target)
        out_a, out_p = [], []
        for a in actions:
            dist = t.distributions.Categorical(probs=a.clamp_min(1e-8))
            out_a.append(dist.sample([1]).view(a.shape[0], 1))
            out_p.append(a)
        return out_a, out_p

    def _criticize(self, agent: int, all_states: Dict, all_actions: Dict,
                   use_target: bool = False):
        net = (
            self.critic_targets[agent]
            if use_target
            else self.critics[agent]
        )
        return safe_return(safe_call(net, all_states, all_actions))

    # ------------------------------------------------------------------
    # storing: one episode list per agent, in lockstep
    # ------------------------------------------------------------------
    def store_episodes(self, episodes: List[List[Union[Dict, Any]]]):
        if len(episodes) != self.agent_num:
            raise ValueError("One episode per agent required.")
        lengths = {len(ep) for ep in episodes}
        if len(lengths) != 1:
            raise ValueError("All agents' episodes must have equal length.")
        for buf, ep in zip(self.replay_buffers, episodes):
            buf.store_episode(ep)

    # ------------------------------------------------------------------
    # updating
    # ------------------------------------------------------------------
    def update(self, update_value=True, update_policy=True,
               update_target=True, concatenate_samples=True, **__):
        """Update every agent (each updates one random ensemble
        member), in parallel over the thread pool."""
        size = min(b.size() for b in self.replay_buffers)
        if size == 0:
            return 0.0, 0.0
        bsize = min(self.batch_size, size)
        indexes = random.sample(range(size), k=bsize)

        def sample_method(buffer, _):
            return len(indexes), [buffer.storage[i] for i in indexes]

        # lockstep batches for every agent
        batches = []
        for buf in self.replay_buffers:
            _, batch = buf.sample_batch(
                -1,
                concatenate_samples,
                sample_method=sample_method,
                sample_attrs=["state", "action", "reward", "next_state",
                              "terminal"],
            )
            batches.append(batch)

        if self.pool_type == "process":
            tasks = []
            for agent in range(self.agent_num):
                visible = self.critic_visible_actors[agent]
                pidx = random.randrange(self.ensemble_size)
                target_actors = [
                    random.choice(self.actor_targets[a]) for a in visible
                ]
                batches_vis = [
                    (batches[a][0], batches[a][1], batches[a][3])
                    for a in visible
                ]
                tasks.append((
                    (
                        self.actors[agent][pidx],
                        self.actor_optims[agent][pidx],
                        self.critics[agent],
                        self.critic_optims[agent],
                        self.critic_targets[agent],
                        target_actors,
                        batches_vis,
                        visible.index(agent),
                        batches[agent][2],
                        batches[agent][4],
                        bsize,
                        self.discount,
                        self.grad_max,
                        update_value,
                        update_policy,
                        self.criterion,
                        (
                            self.action_transform_function,
                            self.action_concat_function,
                            self.state_concat_function,
                            self.reward_function,
                        ),
                    ),
                ))
            results = self.pool.starmap(_process_update_task, tasks)
        else:
            results = self.pool.starmap(
                self._update_agent,
                [
                    (agent, batches, bsize, update_value, update_policy)
                    for agent in range(self.agent_num)
                ],
            )

        if update_target:
            if self.update_rate is not None:
                for agent in range(self.agent_num):
                    for p in range(self.ensemble_size):
                        soft_update(
                            self.actor_targets[agent][p],
                            self.actors[agent][p],
                            self.update_rate,
                        )
                    soft_update(
                        self.critic_targets[agent], self.critics[agent],
                        self.update_rate,
                    )
            else:
                self._update_counter += 1
                if self._update_counter % self.update_steps == 0:
                    for agent in range(self.agent_num):
                        for p in range(self.ensemble_size):
                            hard_update(
                                self.actor_targets[agent][p],
                                self.actors[agent][p],
                            )
                        hard_update(
                            self.critic_targets[agent], self.critics[agent]
                        )

        act_losses = [r[0] for r in results]
        val_losses = [r[1] for r in results]
        return (
            float(np.mean(act_losses)),
            float(np.mean(val_losses)),
        )

    def _update_agent(self, agent: int, batches, bsize: int,
                      update_value: bool, update_policy: bool):
        visible = self.critic_visible_actors[agent]
        policy_idx = random.randrange(self.ensemble_size)
        actor = self.actors[agent][policy_idx]
        actor_optim = self.actor_optims[agent][policy_idx]

        states = [batches[a][0] for a in range(self.agent_num)]
        actions = [batches[a][1] for a in range(self.agent_num)]
        rewards = batches[agent][2]
        next_states = [batches[a][3] for a in range(self.agent_num)]
        terminals = batches[agent][4]

        vis_states = [states[a] for a in visible]
        vis_next_states = [next_states[a] for a in visible]
        vis_actions = [actions[a] for a in visible]

        with t.no_grad():
            next_acts = []
            for a in visible:
                pol = random.choice(self.actor_targets[a])
                next_acts.append(
                    self.action_transform_function(
                        safe_return(safe_call(pol, next_states[a]))
                    )
                )
            all_next_states = self.state_concat_function(vis_next_states)
            all_next_actions = self.action_concat_function(next_acts)
            next_value = self._criticize(
                agent, all_next_states, all_next_actions, use_target=True
            ).view(bsize, 1)
            device = next_value.device
            rew = rewards.to(device).float().view(bsize, 1)
            term = terminals.to(device).float().view(bsize, 1)
            y = self.reward_function(
                rew, self.discount, next_value, term
            )

        all_states = self.state_concat_function(vis_states)
        all_actions = self.action_concat_function(
            [{"action": actions[a]["action"]} for a in visible]
        )
        cur_value = self._criticize(agent, all_states, all_actions).view(
            bsize, 1
        )
        value_loss = self.criterion(cur_value, y.to(cur_value.dtype))
        if update_value:
            self.critic_optims[agent].zero_grad(set_to_none=True)
            value_loss.backward()
            nn.utils.clip_grad_norm_(
                self.critics[agent].parameters(), self.grad_max
            )
            self.critic_optims[agent].step()

        # policy: replace THIS agent's action with its current output
        cur_action = self.action_transform_function(
            safe_return(safe_call(actor, states[agent]))
        )
        policy_actions = []
        for a in visible:
            if a == agent:
                policy_actions.append(cur_action)
            else:
                policy_actions.append({"action": actions[a]["action"]})
        act_value = self._criticize(
            agent,
            self.state_concat_function(vis_states),
            self.action_concat_function(policy_actions),
        )
        act_policy_loss = -act_value.mean()
        if update_policy:
            actor_optim.zero_grad(set_to_none=True)
            act_policy_loss.backward()
            nn.utils.clip_grad_norm_(actor.parameters(), self.grad_max)
            actor_optim.step()

        return (
            -float(act_policy_loss.detach().item()),
            float(value_loss.detach().item()),
        )

    def update_lr_scheduler(self):
        for sch in self.lr_schedulers:
            sch.step()

    def load(self, model_dir, network_map=None, version=-1):
        super().load(model_dir, network_map, version)
        with t.no_grad():
            for agent in range(self.agent_num):
                for p in range(self.ensemble_size):
                    hard_update(
                        self.actors[agent][p],
                        self.actor_targets[agent][p],
                    )
                hard_update(
                    self.critics[agent], self.critic_targets[agent]
                )

    # ------------------------------------------------------------------
    # hooks
    # ------------------------------------------------------------------
    @staticmethod
    def action_transform_function(raw_output_action, *_):
        return {"action": raw_output_action}

    @staticmethod
    def action_concat_function(actions: List[Dict], *_) -> Dict:
        """Concatenate visible agents' action dicts along dim 1."""
        keys = actions[0].keys()
        return {
            k: t.cat([a[k] for a in actions], dim=1) for k in keys
        }

    @staticmethod
    def state_concat_function(states: List[Dict], *_) -> Dict:
        keys = states[0].keys()
        return {
            k: t.cat([s[k] for s in states], dim=1) for k in keys
        }

    @staticmethod
    def reward_function(reward, discount, next_value, terminal, *_):
        return reward + discount * (1.0 - terminal) * next_value

    # ------------------------------------------------------------------
    @classmethod
    def generate_config(cls, config: Union[Dict[str, Any], Config]):
        default = {
            "frame": "MADDPG",
            "models": ["Actor", "Actor", "Critic", "Critic"],
            "model_args": ((), (), (), ()),
            "model_kwargs": ({}, {}, {}, {}),
            "agent_num": 1,
            "optimizer": "Adam",
            "criterion": "MSELoss",
            "criterion_args": (),
            "criterion_kwargs": {},
            "critic_visible_actors": None,
            "sub_policy_num": 0,
            "batch_size": 100,
            "update_rate": 0.001,
            "update_steps": None,
            "actor_learning_rate": 0.0005,
            "critic_learning_rate": 0.001,
            "discount": 0.99,
            "gradient_max": 1e9,
            "replay_size": 500000,
            "replay_device": "cpu",
            "visualize": False,
            "visualize_dir": "",
        }
        config = config or {}
        data = config.data if isinstance(config, Config) else dict(config)
        frame_config = dict(default)
        frame_config.update(data.get("frame_config", {}))
        data["frame"] = frame_config["frame"]
        data["frame_config"] = frame_config
        return Config(**data)
