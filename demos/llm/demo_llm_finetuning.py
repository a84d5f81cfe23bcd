"""GRPO fine-tuning demo: reasoning reward on an offline prompt set.

Reference parity: demos/llm/demo_llm_finetuning.py (the reference downloads
a HF model + GSM8K; offline here, so a small random-init Llama and a
synthetic arithmetic task are used — the training path is identical to the
8B BASELINE config, just smaller).  On an 8-GPU MI355X node launch with:
  torchrun --nproc-per-node 8 demos/llm/demo_llm_finetuning.py
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

import numpy as np
import torch

from agilerl_amd.algorithms.llm.grpo import GRPO
from agilerl_amd.llm_envs import TokenReasoningGym
from agilerl_amd.training.llm import finetune_llm_reasoning

TINY = dict(
    model_type="llama", vocab_size=256, hidden_size=128, intermediate_size=256,
    num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
    max_position_embeddings=512, pad_token_id=0,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iterations", type=int, default=10)
    p.add_argument("--pop-size", type=int, default=2)
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    pop = GRPO.population(
        args.pop_size, model_config=dict(TINY),
        dtype=torch.bfloat16 if use_cuda else torch.float32,
        lora_config={"r": 8, "lora_alpha": 16},
        group_size=8, lr=5e-4, beta=0.04, max_completion_tokens=16,
        device="cuda:0" if use_cuda else "cpu",
    )
    env = TokenReasoningGym(vocab_size=256, prompt_len=16, data_batch_size=2,
                            group_size=8, seed=0)
    agents, history = finetune_llm_reasoning(
        env, pop, max_steps=args.iterations, evo_steps=5, verbose=True,
    )
    if history:
        print("mean reward per cycle:", [round(float(np.mean(h)), 3) for h in history])


if __name__ == "__main__":
    main()
