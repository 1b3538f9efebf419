"""Install verification: the saturn-amd equivalent of the reference's
``examples/wikitext103/simple-verification.py`` (reference lines 33-111),
runnable WITHOUT a GPU (gloo) and without any dataset download.

Builds a small GPT-J-style task, registers the executor library, profiles
it, deep-copies into a 3-point lr sweep, and orchestrates the sweep to
completion.  On an 8xMI355X node, drop the `tiny` flag for the full
GPT-J-6B run.

    SATURN_LIBRARY_PATH=/tmp/udp_lib python examples/simple_verification.py
"""

import copy
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("SATURN_LIBRARY_PATH", tempfile.mkdtemp(prefix="udp_"))

import torch  # noqa: E402

from saturn_amd import HParams, Task, orchestrate, register, search  # noqa: E402
from saturn_amd.executors import (  # noqa: E402
    DDPExecutor,
    FSDPExecutor,
    PipelineExecutor,
    SpilledExecutor,
)
from saturn_amd.models import (  # noqa: E402
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)

TINY = not torch.cuda.is_available()


def main() -> None:
    if TINY:
        model_kwargs = {"n_layer": 2, "n_embd": 128, "n_head": 4,
                        "vocab_size": 512, "n_ctx": 64, "rotary_dim": 16}
        dl = make_token_dataloader(batch_size=4, seq_len=64, vocab=512,
                                   n_batches=8)
        batch_count, gpu_range, n_gpus = 6, [1, 2], 2
    else:
        model_kwargs = {}
        dl = make_token_dataloader(batch_size=8, seq_len=512, vocab=50400,
                                   n_batches=32)
        batch_count, gpu_range, n_gpus = 100, [1, 2, 4, 8], None

    save_dir = tempfile.mkdtemp(prefix="saturn_models_")
    base = Task(
        lambda kwargs=None: get_gptj_model(model_kwargs),
        dl,
        pretraining_loss,
        HParams(lr=1e-3, batch_count=batch_count),
        gpu_range=gpu_range,
        name="gptj_base",
        save_dir=save_dir,
    )

    register("ddp", DDPExecutor)
    register("fsdp", FSDPExecutor)
    register("pipeline", PipelineExecutor)
    register("spilled", SpilledExecutor)

    search([base], log_level=True, n_gpus=n_gpus)

    # lr sweep: deep-copy the profiled task (strategies carry over),
    # mirroring reference simple-verification.py:94-101
    tasks = []
    for lr in (1e-5, 1e-3, 3e-3):
        t = copy.deepcopy(base)
        t.change_name(f"gptj_lr{lr}")
        t.hparams.lr = lr
        tasks.append(t)

    orchestrate(tasks, log_level=True, interval=60 if TINY else 1000,
                n_gpus=n_gpus)
    for t in tasks:
        assert t.has_ckpt(), t.name
        print(f"{t.name}: done, checkpoint at {t.ckpt_path}")
    print("verification OK")


if __name__ == "__main__":
    main()
