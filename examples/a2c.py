"""Self-contained A2C on CartPole using moolib_amd.

Capability parity with the reference's examples/a2c.py (same cooperative
Broker + Accumulator + EnvPool protocol); uses the built-in pure-python
CartPole (no gym dependency).

Run: python examples/a2c.py
"""
import time

import torch
import torch.nn.functional as F

import moolib_amd
from moolib_amd.envs import CartPoleEnv
from moolib_amd.models.cartpole import CartPoleNet
from moolib_amd.utils.record import log_to_file

ADDRESS = "127.0.0.1:5541"

TOTAL_STEPS = 50000
BATCH_SIZE = 2
ROLLOUT_LENGTH = 64
DISCOUNT = 0.99
LR = 1e-3
BASELINE_COST = 0.005
ENTROPY_COST = 0.0006
ADAM_BETAS = (0.0, 0.99)
ADAM_EPSILON = 3e-7


def a2c_loss(state, action, reward, done, initial_core_state, model):
    (_, logits, baseline), _ = model(state, done, initial_core_state, unroll=True)

    reward = reward[1:]
    done = done[1:]
    logits = logits[:-1]
    bootstrap = baseline[-1]
    baseline = baseline[:-1]
    action = action[:-1]

    T, B = reward.shape
    discount = (~done) * DISCOUNT
    returns = torch.empty(T, B)
    acc = bootstrap.detach()
    for t in range(T - 1, -1, -1):
        acc = reward[t] + acc * discount[t]
        returns[t] = acc

    advantages = returns - baseline
    log_policy = F.log_softmax(logits, dim=-1)
    policy = F.softmax(logits, dim=-1)
    cross_entropy = torch.gather(log_policy, -1, action[..., None]).squeeze(-1)

    pg_loss = -torch.mean(cross_entropy * advantages.detach())
    baseline_loss = 0.5 * torch.mean(advantages**2)
    entropy_loss = torch.mean(policy * log_policy)
    return pg_loss, baseline_loss, entropy_loss


def train(total_steps=TOTAL_STEPS, address=ADDRESS, log=True, seed=1):
    envs = moolib_amd.EnvPool(
        lambda: CartPoleEnv(), num_processes=1, batch_size=BATCH_SIZE, num_batches=1
    )
    torch.manual_seed(seed)
    model = CartPoleNet(obs_dim=4, num_actions=2, use_lstm=True)

    broker = moolib_amd.Broker()
    bound = broker.listen(address)
    connect_addr = [a for a in bound if a.startswith("tcp://127")] or bound
    accumulator = moolib_amd.Accumulator("a2c", model.parameters(), model.buffers())
    accumulator.connect(connect_addr[0])

    opt = torch.optim.Adam(model.parameters(), lr=LR, betas=ADAM_BETAS, eps=ADAM_EPSILON)

    states, actions, rewards, dones, core_states = [], [], [], [], []
    episode_returns = []
    action_t = torch.zeros(BATCH_SIZE, dtype=torch.int64)
    episode_step_t = torch.zeros(BATCH_SIZE, dtype=torch.int64)
    episode_return_t = torch.zeros(BATCH_SIZE)
    core_state_t = model.initial_state(batch_size=BATCH_SIZE)

    local_loss_computes = 0
    local_update_steps = 0
    recent_returns = []

    while True:
        broker.update()
        accumulator.update()
        if not accumulator.connected():
            time.sleep(0.25)
            continue

        obs = envs.step(0, action_t).result()
        episode_step_t += 1
        episode_return_t += obs["reward"]
        core_states.append(core_state_t)

        state_t = obs["state"].to(torch.float, copy=True)
        done_t = obs["done"].clone()
        with torch.no_grad():
            (action_t, _, _), core_state_t = model(state_t, done_t, core_state_t)

        episode_returns.append(episode_return_t.clone())
        if done_t.any():
            for r in episode_return_t[done_t].tolist():
                recent_returns.append(r)
            recent_returns = recent_returns[-50:]
        episode_step_t *= ~done_t
        episode_return_t *= ~done_t

        states.append(state_t)
        actions.append(action_t)
        rewards.append(obs["reward"].clone())
        dones.append(done_t)

        if accumulator.wants_state():
            accumulator.set_state({"optimizer": opt.state_dict()})
        if accumulator.has_new_state():
            opt.load_state_dict(accumulator.state()["optimizer"])

        if accumulator.wants_gradients():
            if len(states) < ROLLOUT_LENGTH + 1:
                accumulator.skip_gradients()
            else:
                state = torch.stack(states)
                action = torch.stack(actions)
                reward = torch.stack(rewards)
                done = torch.stack(dones)
                pg_loss, baseline_loss, entropy_loss = a2c_loss(
                    state, action, reward, done, core_states[0], model
                )
                del states[:-1], actions[:-1], rewards[:-1], dones[:-1]
                del episode_returns[:-1], core_states[:-1]

                loss = pg_loss + BASELINE_COST * baseline_loss + ENTROPY_COST * entropy_loss
                loss.backward()
                local_loss_computes += 1
                agent_steps = local_loss_computes * ROLLOUT_LENGTH * BATCH_SIZE
                if agent_steps >= total_steps:
                    return recent_returns

                torch.nn.utils.clip_grad_norm_(model.parameters(), 100)
                if log and local_loss_computes % 20 == 0:
                    mean_ret = (
                        sum(recent_returns) / len(recent_returns) if recent_returns else float("nan")
                    )
                    print(
                        "loss computes %d (%d steps), mean episode return %.1f, updates %d"
                        % (local_loss_computes, agent_steps, mean_ret, local_update_steps)
                    )
                    log_to_file(
                        step=agent_steps,
                        mean_episode_return=mean_ret,
                        pg_loss=pg_loss.item(),
                    )
                accumulator.reduce_gradients(BATCH_SIZE)

        if accumulator.has_gradients():
            opt.step()
            accumulator.zero_gradients()
            local_update_steps += 1


if __name__ == "__main__":
    train()
