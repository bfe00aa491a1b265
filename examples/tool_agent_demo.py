"""Agentic tool use with GUARANTEED-well-formed calls.

The agent registers two @skill functions; `app.ai(tools=True)` runs the
tool loop: each round the constrained decoder can ONLY emit a JSON call
matching one of the skill signatures (or the built-in final_answer), the
skill executes locally, and its result feeds the next round.  Works on
CPU with the tiny model (random weights — the point is the mechanics and
the guarantee, not the answers).

    PYTHONPATH=. python examples/tool_agent_demo.py
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

from agentfield_amd.engine import LLMEngine
from agentfield_amd.models import CONFIGS
from agentfield_amd.sdk import Agent
from agentfield_amd.sdk.ai import ByteTokenizer, EngineRunner, set_runner


def main():
    device = "cuda" if torch.cuda.is_available() else "cpu"
    name = "llama-3-8b" if device == "cuda" else "tiny"
    kw = {} if device == "cuda" else {"dtype": torch.float32,
                                      "num_pages": 256, "page_size": 4,
                                      "enable_graphs": False}
    eng = LLMEngine(CONFIGS[name], device=device, max_num_seqs=4, seed=1,
                    **kw)
    runner = EngineRunner(eng, ByteTokenizer(CONFIGS[name].vocab_size))
    set_runner(name, runner)

    app = Agent("calculator", auto_register=False)

    @app.skill()
    def add(a: int = 0, b: int = 0):
        print(f"  [skill] add({a}, {b})")
        return {"sum": a + b}

    @app.skill()
    def lookup(city: str = ""):
        print(f"  [skill] lookup({city!r})")
        return {"population": {"tokyo": 37_000_000}.get(city.lower(),
                                                        "unknown")}

    print("running ai(tools=True) — every tool call below is")
    print("schema-constrained: a malformed call is IMPOSSIBLE\n")
    answer = app.ai("What is 2+3, and how many people live in Tokyo?",
                    tools=True, max_tool_rounds=3, model=name,
                    max_tokens=64, temperature=0.9)
    print(f"\nfinal answer (random-weight model, mechanics demo): "
          f"{answer!r}")
    runner.shutdown()


if __name__ == "__main__":
    main()
