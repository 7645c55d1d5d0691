import sys
sys.path.insert(0, ".")
import torch
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig

torch.manual_seed(17)
cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64, block_size=16,
                      max_model_len=128, device="cuda:0")
eng = LlmEngine(cfg); eng.start()
t_short = "embedding parity probe"
t_long = "second, longer text for the batch"
prompts = [eng.tokenizer.encode(t) for t in (t_short, t_long)]
print("lens:", [len(p) for p in prompts])
plan = {"mode": "embed", "prompts": prompts}
# replicate _exec_embed but dump the raw hidden
import torch as T
b = 2; lens = [len(p) for p in prompts]; smax = max(lens); dev = eng.device
tokens = T.zeros(b, smax, dtype=T.long); positions = T.zeros(b, smax, dtype=T.int32)
for i in range(b):
    tokens[i, :lens[i]] = T.tensor(prompts[i]); positions[i, :lens[i]] = T.arange(lens[i], dtype=T.int32)
attn_ctx = {"mode": "prefill", "batch": b, "seq": smax,
            "seq_lens": T.tensor(lens, dtype=T.int32, device=dev)}
with torch.inference_mode():
    hid = eng.model(tokens.view(-1).to(dev), positions.view(-1).to(dev),
                    kv_caches=None, attn_ctx=attn_ctx, return_hidden=True)
hid = hid.view(b, smax, -1).float()
for i in range(b):
    nan_rows = torch.isnan(hid[i]).any(-1)
    print(f"row {i}: nan token-rows = {nan_rows.nonzero().flatten().tolist()}")
    print(f"        valid-rows finite: {torch.isfinite(hid[i, :lens[i]]).all().item()}")
