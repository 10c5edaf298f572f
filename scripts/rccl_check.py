"""RCCL smoke: init torch.distributed with the nccl(=RCCL) backend under
torchrun exactly like the driver's scale run, run the collectives bench.py
uses (barrier, all_reduce MAX, broadcast, bucketed all_reduce at fp32)
and a DistributedGrads round-trip. World size comes from torchrun."""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch
from code_intelligence_amd.parallel.ddp import (init_distributed,
                                                broadcast_parameters,
                                                DistributedGrads)
from code_intelligence_amd.models.awd_lstm import AWDLSTM

rank, world = init_distributed()
assert torch.distributed.is_initialized(), "expected torchrun env"
assert torch.distributed.get_backend() == "nccl", \
    torch.distributed.get_backend()
dev = torch.device("cuda", 0)
t = torch.full((1024,), float(rank + 1), device=dev)
torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
assert float(t[0]) == world
torch.distributed.barrier()
m = AWDLSTM(vocab_sz=512, emb_sz=32, n_hid=48, n_layers=2
            ).to(dev, torch.bfloat16)
broadcast_parameters(m)
dg = DistributedGrads(m, bucket_mb=1.0)
m.train(); m.reset(4)
x = torch.randint(0, 512, (4, 8), device=dev)
dg.prepare()
from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
raw, outs = m.encoder(x)
loss = tied_decoder_ce(outs[-1].reshape(-1, 32), m.decoder.decoder.weight,
                       m.decoder.decoder.bias, x.reshape(-1))
loss.backward()
dg.finalize()
for p in m.parameters():
    if p.grad is not None:
        assert torch.isfinite(p.grad.float()).all()
torch.cuda.synchronize()
if rank == 0:
    print(f"rccl ok: world={world} backend=nccl loss={float(loss):.3f}")
torch.distributed.destroy_process_group()
