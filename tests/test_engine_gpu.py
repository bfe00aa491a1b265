"""GPU engine tests: graph-captured decode vs eager, and a small E2E run."""
import pytest
import torch

from agentfield_amd.engine import LLMEngine, SamplingParams
from agentfield_amd.models import CONFIGS

pytestmark = pytest.mark.gpu


def test_graph_decode_matches_eager():
    cfg = CONFIGS["tiny"]
    prompts = [[1, 5, 9, 20, 7], [3, 7, 11], [2, 4, 6, 8]]
    sp = SamplingParams(max_tokens=8, ignore_eos=True)
    outs = {}
    for graphs in (False, True):
        eng = LLMEngine(cfg, device="cuda", page_size=4, num_pages=128,
                        max_num_seqs=4, enable_graphs=graphs, seed=3)
        outs[graphs] = eng.generate(prompts, sp)
        del eng
        torch.cuda.empty_cache()
    assert outs[False] == outs[True], "hipGraph decode diverges from eager"


def test_debug1b_generates():
    cfg = CONFIGS["debug-1b"]
    eng = LLMEngine(cfg, device="cuda", num_pages=512, max_num_seqs=8,
                    enable_graphs=True, seed=1)
    torch.manual_seed(0)
    prompts = [torch.randint(0, cfg.vocab_size, (64,)).tolist() for _ in range(4)]
    outs = eng.generate(prompts, SamplingParams(max_tokens=16, ignore_eos=True))
    assert all(len(o) == 16 for o in outs)
    assert all(0 <= t < cfg.vocab_size for o in outs for t in o)
    m = eng.metrics
    assert m["prefill_tokens"] == 256 and m["decode_tokens"] >= 4 * 15
