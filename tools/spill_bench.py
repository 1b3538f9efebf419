"""Spill engine probe: seconds/batch of the spilled executor (world-1
ZeRO-3 + host offload) across partition counts, on a model big enough for
the H2D stream to matter."""
import os, sys, tempfile
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from saturn_amd import HParams, Task
from saturn_amd.executors.spilled import _run_spilled
from saturn_amd.models.gptj import get_gptj_model, make_token_dataloader, pretraining_loss

kw = {"n_layer": 12}  # 2.7B params, ~5.4 GB bf16
t = Task(lambda kwargs=None: get_gptj_model(kw),
         make_token_dataloader(batch_size=8, seq_len=512, vocab=50400, n_batches=8),
         pretraining_loss, HParams(lr=1e-4, batch_count=8),
         name="spill_probe", save_dir=tempfile.mkdtemp())
for parts in (1, 2, 4, 12):
    bt = _run_spilled(t, 0, {"partitions": parts, "offload": True}, True)
    print(f"partitions={parts}: {bt*1000:.1f} ms/batch", flush=True)
