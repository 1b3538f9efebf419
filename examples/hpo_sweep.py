"""Heterogeneous HPO batch: the saturn-amd equivalent of the reference's
``examples/wikitext103/WikiText103.py`` (lines 34-106) — a mixed-model,
mixed-batch-size hyperparameter sweep submitted as ONE job batch, profiled
once, then gang-scheduled to completion.

Runs tiny on CPU (gloo world 2) with no downloads; on an 8xMI355X node the
full-scale branch sweeps GPT-J-6B at the reference's batch sizes {8,16,32}
alongside a GPT-2-XL job.

    SATURN_LIBRARY_PATH=/tmp/udp_lib python examples/hpo_sweep.py
"""

import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("SATURN_LIBRARY_PATH", tempfile.mkdtemp(prefix="udp_"))

import torch  # noqa: E402

from saturn_amd import HParams, Task, orchestrate, register, search  # noqa: E402
from saturn_amd.executors import DDPExecutor, FSDPExecutor  # noqa: E402
from saturn_amd.models import (  # noqa: E402
    get_gpt2_model,
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)

TINY = not torch.cuda.is_available()


def main() -> None:
    save_dir = tempfile.mkdtemp(prefix="saturn_models_")
    register("ddp", DDPExecutor)
    register("fsdp", FSDPExecutor)

    tasks = []
    if TINY:
        gptj_kw = {"n_layer": 2, "n_embd": 128, "n_head": 4,
                   "vocab_size": 512, "n_ctx": 64, "rotary_dim": 16}
        gpt2_kw = {"preset": "small", "n_layer": 2, "n_ctx": 64}
        for bs, lr in ((2, 1e-4), (4, 3e-4)):
            tasks.append(Task(
                lambda kwargs=None, kw=gptj_kw: get_gptj_model(kw),
                make_token_dataloader(batch_size=bs, seq_len=64, vocab=512,
                                      n_batches=8),
                pretraining_loss,
                HParams(lr=lr, batch_count=6),
                gpu_range=[1, 2],
                name=f"gptj_bs{bs}_lr{lr}",
                save_dir=save_dir,
            ))
        tasks.append(Task(
            lambda kwargs=None, kw=gpt2_kw: get_gpt2_model(kw),
            make_token_dataloader(batch_size=2, seq_len=64, vocab=50257,
                                  n_batches=8),
            pretraining_loss,
            HParams(lr=1e-4, batch_count=6),
            gpu_range=[1],
            name="gpt2_tiny",
            save_dir=save_dir,
        ))
        n_gpus, interval = 2, 60
    else:
        # reference WikiText103.py sweeps GPT-J at batch sizes 8..32
        for bs in (8, 16, 32):
            tasks.append(Task(
                lambda kwargs=None: get_gptj_model({}),
                make_token_dataloader(batch_size=bs, seq_len=512,
                                      vocab=50400, n_batches=64),
                pretraining_loss,
                HParams(lr=1e-5, batch_count=200),
                gpu_range=[1, 2, 4, 8],
                name=f"gptj6b_bs{bs}",
                save_dir=save_dir,
            ))
        tasks.append(Task(
            lambda kwargs=None: get_gpt2_model({"preset": "xl"}),
            make_token_dataloader(batch_size=16, seq_len=512, vocab=50257,
                                  n_batches=64),
            pretraining_loss,
            HParams(lr=1e-4, batch_count=200),
            gpu_range=[1, 2, 4],
            name="gpt2xl_bs16",
            save_dir=save_dir,
        ))
        n_gpus, interval = None, 1000

    search(tasks, log_level=True, n_gpus=n_gpus)
    orchestrate(tasks, log_level=True, interval=interval, n_gpus=n_gpus)
    for t in tasks:
        assert t.has_ckpt(), t.name
        print(f"{t.name}: done, checkpoint at {t.ckpt_path}")
    print("HPO sweep OK")


if __name__ == "__main__":
    main()
