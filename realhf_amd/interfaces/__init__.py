from realhf_amd.interfaces import dpo, gen, grpo, ppo, rw, sft  # noqa: F401
