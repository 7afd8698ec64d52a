"""Family scaffolding generator test (mirror reference template codegen)."""
import importlib
import sys

import torch


def test_generated_family_decodes(tmp_path):
    from bloombee_amd.models.template import generate_family

    d = generate_family(
        "bloombee_amd/models/template/spec_llama.yaml", str(tmp_path))
    sys.path.insert(0, str(tmp_path))
    try:
        importlib.import_module("myllama")
        from bloombee_amd.engine import LocalEngine

        eng = LocalEngine("myllama-tiny", device="cpu", seed=0,
                          kv_max_tokens=4096)
        ids = torch.randint(0, 900, (1, 6),
                            generator=torch.Generator().manual_seed(2))
        out = eng.generate_greedy(ids, 4)
        assert out.shape == (1, 4)
        # decode == prefill consistency on the generated block
        kv1 = eng.kv_pool.allocate(1, 64)
        full = eng.prefill(ids, kv1)
        kv1.close()
        kv2 = eng.kv_pool.allocate(1, 64)
        tok = eng.prefill(ids[:, :1], kv2)
        for t in range(1, 6):
            tok = eng.decode_step(ids[0, t].view(1), kv2)
        kv2.close()
        assert torch.equal(full, tok)
    finally:
        sys.path.remove(str(tmp_path))
