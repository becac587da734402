import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from substratus_amd.models import CausalLM, get_config
from substratus_amd.serve.engine import Engine, EngineConfig

torch.manual_seed(21)
mcfg = get_config("llama-tiny")
model = CausalLM(mcfg, device="cuda", dtype=torch.bfloat16)
model.init_random(seed=9)
for use_graphs in (False, True):
    eng = Engine(EngineConfig(model="llama-tiny", dtype=torch.bfloat16,
                              max_batch=4, max_seq_len=128, kv_blocks=64,
                              use_graphs=use_graphs), device="cuda", model=model)
    sids = [eng.add_request([5, 9, 2, 7, 1, 3] * 4, max_new_tokens=12),
            eng.add_request([8, 1, 4], max_new_tokens=12)]
    toks = {sid: [] for sid in sids}
    n = 0
    while eng.has_work() and n < 40:
        for o in eng.step():
            toks[o.seq_id].append(o.token)
        n += 1
    print("graphs" if use_graphs else "eager ", [toks[s] for s in sids],
          "free", eng.kv.allocator.free_count)

# 7b timing graph vs eager
for use_graphs in (True, False):
    cfg = EngineConfig(model="llama2-7b", dtype=torch.bfloat16, max_batch=64,
                       max_seq_len=1024, use_graphs=use_graphs)
    eng = Engine(cfg, device="cuda")
    g = torch.Generator().manual_seed(7)
    for _ in range(64):
        eng.add_request(torch.randint(0, 32000, (512,), generator=g).tolist(),
                        max_new_tokens=400)
    while eng.waiting:
        eng.step()
    for _ in range(8):
        eng.decode_only_step()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(32):
        eng.decode_only_step()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / 32
    print(f"7b decode use_graphs={use_graphs}: {dt*1000:.2f} ms/step "
          f"({64/dt:.0f} tok/s)")
