"""Op dispatch: gfx950 HIP kernels on device, pure-torch reference on CPU.

Policy (driver contract): on a GPU box the HIP extension MUST be the path
that runs — if a tensor is on device and the extension is missing, we raise
instead of silently falling back to eager PyTorch.
"""
from bloombee_amd.ops.interface import (  # noqa: F401
    HAVE_HIP_OPS,
    alibi_slopes_for,
    attn_decode,
    attn_paged,
    attn_paged_mixed,
    attn_paged_qkv,
    attn_prefill,
    fuse_norm_linear_ok,
    gelu_tanh,
    hip_ops,
    kv_gather,
    kv_write,
    layer_norm,
    linear,
    linear_w4,
    mfma_selftest,
    moe_gemm_grouped,
    quant4_pack,
    quant4_unpack,
    rms_norm,
    rms_norm_residual,
    rope_apply_,
    rope_kv_write_,
    rope_cos_sin,
    set_attn_sparsity,
    swiglu,
)
