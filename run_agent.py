"""Eval CLI — API-compatible with the reference ``run_agent.py``
(flags --run/--episodes/--headless/--random, reference run_agent.py:51-59):
loads the actor from an MLflow run's artifacts and rolls out N episodes,
deterministic by default.
"""

import logging
from argparse import ArgumentParser, Namespace
from itertools import count
from pathlib import Path

import torch
import tqdm

from torch_actor_critic_amd import envs
from torch_actor_critic_amd.envs.visual import MultiObservation
from torch_actor_critic_amd.utils import checkpoint as ckpt

logging.basicConfig(level=logging.INFO)
logger = logging.getLogger(__name__)


def run_agent(actor, env, episodes: int, deterministic: bool = True,
              render: bool = True, device=None):
    """Roll out episodes (reference run_agent.py:19-48)."""
    device = device or next(actor.parameters()).device
    all_returns = []
    for e in range(episodes):
        metrics = {"ep_ret": 0.0, "ep_len": 0}
        state = env.reset()
        pbar = tqdm.tqdm(count(), desc=f"Episode {e}", ncols=0)
        for _ in pbar:
            with torch.no_grad():
                if isinstance(state, MultiObservation):
                    s = MultiObservation(state.features.to(device),
                                         state.frame.to(device))
                else:
                    s = torch.as_tensor(state, dtype=torch.float32,
                                        device=device)
                action, _ = actor(s, deterministic=deterministic)
            state, reward, done, _ = env.step(action.detach().cpu().numpy())
            metrics["ep_len"] += 1
            metrics["ep_ret"] += float(reward)
            pbar.set_postfix(metrics)
            if render:
                env.render()
            if done:
                break
        all_returns.append(metrics["ep_ret"])
    return all_returns


def parse_arguments() -> Namespace:
    parser = ArgumentParser("Soft Actor-Critic evaluator (MI355X-native).")
    parser.add_argument("--run", type=str, help="MLflow run id to load")
    parser.add_argument("--episodes", type=int, default=100,
                        help="Number of test episodes")
    parser.add_argument("--headless", action="store_false", dest="render",
                        help="Disable rendering")
    parser.add_argument("--random", action="store_false",
                        dest="deterministic", help="Stochastic policy")
    parser.add_argument("--device", default=None)
    return parser.parse_args()


def main():
    args = parse_arguments()
    artifact_path = Path("mlruns", "0", args.run, "artifacts")

    sac_params = ckpt.get_run_params(args.run)
    environment = sac_params.get("environment", "Humanoid-v2")
    env = envs.make(environment)

    device = torch.device(args.device) if args.device else (
        torch.device("cuda") if torch.cuda.is_available()
        else torch.device("cpu"))
    actor = ckpt.load_model(str(artifact_path / "actor")).to(device)
    actor.eval()

    run_agent(actor, env, args.episodes, args.deterministic, args.render,
              device)


if __name__ == "__main__":
    main()
