"""Shared full-training harnesses (reference CI gates: CartPole
smoothed reward > 150, Pendulum > -400, 5 consecutive episodes)."""
import torch as t

from machin_amd.env.envs.classic_control import CartPoleEnv, PendulumEnv


def train_cartpole(
    framework,
    act_fn,
    store_fn=None,
    update_fn=None,
    max_episodes=600,
    target=150.0,
    wins_needed=5,
    update_every_episode=True,
    warmup_size=500,
):
    """Generic CartPole trainer. act_fn(state_tensor) -> int action.
    Returns True when solved."""
    env = CartPoleEnv(seed=0)
    smoothed, wins = 0.0, 0
    for _ in range(max_episodes):
        obs = t.tensor(env.reset(), dtype=t.float32).view(1, 4)
        total_reward = 0.0
        transitions = []
        done = False
        while not done:
            with t.no_grad():
                action = act_fn(obs)
            obs_next, reward, done, _ = env.step(int(action))
            obs_next = t.tensor(obs_next, dtype=t.float32).view(1, 4)
            total_reward += reward
            transitions.append(
                {
                    "state": {"state": obs},
                    "action": {"action": t.tensor([[int(action)]])},
                    "next_state": {"state": obs_next},
                    "reward": reward,
                    "terminal": done and env.steps < env.max_episode_steps,
                }
            )
            obs = obs_next
        if store_fn is not None:
            store_fn(transitions)
        else:
            framework.store_episode(transitions)
        if update_fn is not None:
            update_fn(transitions)
        elif update_every_episode:
            if framework.replay_buffer.size() > warmup_size:
                for _ in range(min(len(transitions), 50)):
                    framework.update()
        smoothed = smoothed * 0.9 + total_reward * 0.1
        if smoothed > target:
            wins += 1
            if wins >= wins_needed:
                return True
        else:
            wins = 0
    return False


def train_pendulum(
    framework,
    act_fn,
    max_episodes=400,
    target=-400.0,
    wins_needed=5,
    warmup_size=500,
    updates_per_episode=100,
):
    env = PendulumEnv(seed=0)
    smoothed, wins = -1600.0, 0
    for _ in range(max_episodes):
        obs = t.tensor(env.reset(), dtype=t.float32).view(1, 3)
        total_reward = 0.0
        transitions = []
        done = False
        while not done:
            with t.no_grad():
                action = act_fn(obs)
            obs_next, reward, done, _ = env.step(action.view(-1).numpy())
            obs_next = t.tensor(obs_next, dtype=t.float32).view(1, 3)
            total_reward += reward
            transitions.append(
                {
                    "state": {"state": obs},
                    "action": {"action": action.view(1, -1)},
                    "next_state": {"state": obs_next},
                    "reward": reward / 10.0,
                    "terminal": False,
                }
            )
            obs = obs_next
        framework.store_episode(transitions)
        if framework.replay_buffer.size() > warmup_size:
            for _ in range(updates_per_episode):
                framework.update()
        smoothed = smoothed * 0.9 + total_reward * 0.1
        if smoothed > target:
            wins += 1
            if wins >= wins_needed:
                return True
        else:
            wins = 0
    return False
