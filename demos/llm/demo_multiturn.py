"""Multi-turn LLM fine-tuning demo: GRPO on a token guessing game.

Reference parity: the reference's multi-turn flow
(finetune_llm_multiturn, training/llm/multiturn.py:43) on its multi-turn
token env — here with the first-party SyncMultiTurnVecEnv and a tiny
random-init Llama (offline; the training path scales to real models by
swapping model_config for model_name_or_path).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import torch

from agilerl_amd.algorithms.llm.grpo import GRPO
from agilerl_amd.llm_envs.multiturn import SyncMultiTurnVecEnv, TokenGuessEnv
from agilerl_amd.training.llm.multiturn import finetune_llm_multiturn

TINY = dict(
    model_type="llama", vocab_size=128, hidden_size=64, intermediate_size=128,
    num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
    max_position_embeddings=256, pad_token_id=0,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iterations", type=int, default=6)
    p.add_argument("--pop-size", type=int, default=2)
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    pop = GRPO.population(
        args.pop_size, model_config=dict(TINY),
        dtype=torch.bfloat16 if use_cuda else torch.float32,
        lora_config={"r": 8, "lora_alpha": 16},
        group_size=4, lr=1e-3, max_completion_tokens=4,
        device="cuda:0" if use_cuda else "cpu",
    )
    env = SyncMultiTurnVecEnv(
        lambda: TokenGuessEnv(128, prompt_len=6, max_turns=2),
        data_batch_size=2, group_size=4, max_turns=2, seed=0,
    )
    agents, hist = finetune_llm_multiturn(
        env, pop, max_steps=args.iterations, evo_steps=3, verbose=True,
    )
    print(f"final best fitness: {max(a.fitness[-1] for a in agents):.3f}")


if __name__ == "__main__":
    main()
