"""7B GPT-shaped config at seq 4096 — 288 GB HBM sizing stressor
(BASELINE.json config 5; not present in the reference)."""
from midgpt_amd.config import ExperimentConfig, GPTConfig

config = ExperimentConfig(
    rundir='',
    data_dir='data/openwebtext',
    learning_rate=3e-4,
    batch_size=128,
    warmup_steps=2000,
    min_lr=3e-5,
    lr_decay_steps=25_000,
    max_steps=25_000,
    beta2=0.95,
    weight_decay=1e-4,
    eval_interval=1000,
    compute_dtype='bfloat16',
    param_dtype='float32',
    g_accum_iters=1,
    shard_model=True,
    model_config=GPTConfig(
        block_size=4096, vocab_size=50304, n_layer=32, n_head=32, n_embd=4096, dropout=0.0),
)
