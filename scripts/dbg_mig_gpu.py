import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
from resilient_llm_amd.models import LlamaForCausalLM, get_config
from resilient_llm_amd.engine.graph import install_graph_runner

cfg = get_config("tiny-128")
m1 = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=7)
m2 = LlamaForCausalLM(cfg, device="cuda:0", dtype=torch.bfloat16, seed=7)

def eng(m, graphs=False):
    kv = PagedKVCache.for_model(cfg, 128, device="cuda:0")
    e = LLMEngine(m, kv, max_batch_size=4)
    if graphs:
        install_graph_runner(e)
    return e

for temp, graphs in ((0.0, False), (0.8, False), (0.8, True)):
    params = lambda: SamplingParams(max_tokens=16, temperature=temp, seed=5, stop_on_eos=False)
    prompt = list(range(11, 64))
    ref = eng(m1); ref.add_request("g", prompt, params())
    want = []
    while ref.has_work():
        for o in ref.step(): want.append(o.token_id)
    src = eng(m1); dst = eng(m2, graphs=graphs)
    src.add_request("g", prompt, params())
    got = []
    for _ in range(7):
        for o in src.step(): got.append(o.token_id)
    nmig = len(got)
    src.request_extract("g"); src.step()
    state = src.take_extracted("g")
    dst.queue_adopt(state)
    for _ in range(200):
        if not dst.has_work(): break
        for o in dst.step(): got.append(o.token_id)
    div = next((i for i in range(min(len(got), len(want))) if got[i] != want[i]), None)
    print(f"temp={temp} graphs={graphs} nmig={nmig} match={got == want} first_div={div} want={want} got={got}", flush=True)
