"""Profiling harness: GRPO learn step (tiny model by default).

Run under rocprofv3 for kernel stats:
  rocprofv3 --kernel-trace --stats -- python demos/profile_grpo.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from agilerl_amd.algorithms.llm.grpo import GRPO

TINY = dict(model_type="llama", vocab_size=2048, hidden_size=256, intermediate_size=512,
            num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=4,
            max_position_embeddings=2048, pad_token_id=0)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--steps", type=int, default=3)
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    agent = GRPO(model_config=dict(TINY), dtype=dtype, lora_config={"r": 16},
                 group_size=8, micro_batch_size=8, beta=0.04, device=device)
    import numpy as np

    for _ in range(args.steps):
        ids = torch.randint(1, 2048, (args.batch, args.seq_len), device=device)
        P = args.seq_len // 2
        pos = torch.arange(args.seq_len - 1, device=device).unsqueeze(0)
        mask = (pos + 1 >= P).float().expand(args.batch, args.seq_len - 1)
        rewards = torch.rand(args.batch)
        stats = agent.learn({"ids": ids, "action_mask": mask, "rewards": rewards})
        print(stats)


if __name__ == "__main__":
    main()
