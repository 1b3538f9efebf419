"""Expert-parallel MoE training through the full Saturn pipeline:
register the EP technique, profile it against DDP, gang-schedule.

Runs tiny on CPU (gloo world 2, experts sharded 2+2); on an 8xMI355X node
the full branch trains the Mixtral 8x7B-proxy with experts sharded across
the gang and tokens exchanged over xGMI all-to-all.

    SATURN_LIBRARY_PATH=/tmp/udp_lib python examples/moe_training.py
"""

import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("SATURN_LIBRARY_PATH", tempfile.mkdtemp(prefix="udp_"))

import torch  # noqa: E402

from saturn_amd import HParams, Task, orchestrate, register, search  # noqa: E402
from saturn_amd.executors import DDPExecutor, ExpertParallelExecutor  # noqa: E402
from saturn_amd.models.gptj import make_token_dataloader  # noqa: E402
from saturn_amd.models.mixtral import get_mixtral_model, mixtral_loss  # noqa: E402

TINY = not torch.cuda.is_available()


def main() -> None:
    save_dir = tempfile.mkdtemp(prefix="saturn_models_")
    register("ddp", DDPExecutor)
    register("expert", ExpertParallelExecutor)

    if TINY:
        kw = {"n_layer": 2, "n_embd": 64, "n_head": 2, "n_kv_head": 1,
              "vocab_size": 128, "n_ctx": 32, "ffn_dim": 96, "n_expert": 4,
              "top_k": 2}
        dl = make_token_dataloader(batch_size=4, seq_len=32, vocab=128,
                                   n_batches=8)
        batch_count, gpu_range, n_gpus, interval = 6, [1, 2], 2, 60
    else:
        kw = {"preset": "8x7b-proxy"}
        dl = make_token_dataloader(batch_size=8, seq_len=2048, vocab=32000,
                                   n_batches=32)
        batch_count, gpu_range, n_gpus, interval = 100, [2, 4, 8], None, 1000

    t = Task(
        lambda kwargs=None, kw=kw: get_mixtral_model(kw),
        dl,
        mixtral_loss,
        HParams(lr=1e-4, batch_count=batch_count),
        gpu_range=gpu_range,
        name="mixtral_moe",
        save_dir=save_dir,
    )
    search([t], log_level=True, n_gpus=n_gpus)
    chosen = {g: s.executor.name for g, s in t.strategies.items()
              if s is not None and s.feasible}
    print("feasible strategies:", chosen)
    orchestrate([t], log_level=True, interval=interval, n_gpus=n_gpus)
    assert t.has_ckpt()
    print(f"done, checkpoint at {t.ckpt_path}")


if __name__ == "__main__":
    main()
