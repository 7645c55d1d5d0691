import asyncio, sys
sys.path.insert(0, ".")
import torch
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig

def run(c):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(c)
    finally:
        for t in asyncio.all_tasks(loop): t.cancel()
        loop.run_until_complete(asyncio.sleep(0)); loop.close()

def mk(device):
    torch.manual_seed(17)
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                          block_size=16, max_model_len=128, device=device)
    e = LlmEngine(cfg); e.start(); return e

t_short = "embedding parity probe"
t_long = "second, longer text for the batch"
g = mk("cuda:0")
gb = torch.tensor(run(g.embed_batch([t_short, t_long])))
gs0 = torch.tensor(run(g.embed_batch([t_short])))
gs1 = torch.tensor(run(g.embed_batch([t_long])))
grev = torch.tensor(run(g.embed_batch([t_long, t_short])))
print("gpu batched vs solo short:", float((gb[0]*gs0[0]).sum()))
print("gpu batched vs solo long :", float((gb[1]*gs1[0]).sum()))
print("gpu reversed short       :", float((grev[1]*gs0[0]).sum()))
c = mk("cpu")
cb = torch.tensor(run(c.embed_batch([t_short, t_long])))
print("cpu vs gpu solo short    :", float((cb[0]*gs0[0]).sum()))
print("cpu vs gpu solo long     :", float((cb[1]*gs1[0]).sum()))
