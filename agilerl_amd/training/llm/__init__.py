"""LLM fine-tuning workload loops (GRPO-family reasoning, SFT, DPO, multiturn)."""


def run_llm_workload(trainer, workload: str):
    if workload == "llm_reasoning":
        from .reasoning import finetune_llm_reasoning

        return finetune_llm_reasoning(trainer)
    if workload == "llm_sft":
        from .sft import finetune_llm_sft

        return finetune_llm_sft(trainer)
    if workload == "llm_preference":
        from .preference import finetune_llm_preference

        return finetune_llm_preference(trainer)
    raise NotImplementedError(workload)
