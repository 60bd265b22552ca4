from midgpt_amd.models.gpt import GPT, count_params  # noqa: F401
