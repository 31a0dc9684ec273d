"""adanet_amd.ops — CDNA4 kernel-backed ops with CPU fp32 reference paths.

Kernel inventory:
  K1  gemm_nt_bf16 / HipLinear          (csrc/gemm.hip)
  K2  softmax_xent                       (csrc/softmax_xent.hip)
  K3  HipDropout / relu epilogue+bwd     (csrc/elementwise.hip)
  K4  FusedSGD / FusedAdam               (csrc/optim.hip)
  K5  weighted_sum_logits mixer          (csrc/mixer.hip)
  K6  L1/complexity (folded into weights' autograd + mixer bwd)
  K8  HipLayerNorm                       (csrc/layernorm.hip)
  K10 colsum / argmax_correct metrics    (csrc/reduce.hip)
  K11 best-candidate mux: host-side (engine)
"""

from adanet_amd.ops import _extension
from adanet_amd.ops.dropout import HipDropout
from adanet_amd.ops.layernorm import HipLayerNorm
from adanet_amd.ops.linear import HipLinear, gemm_nt, transpose2d
from adanet_amd.ops.mixer import weighted_sum_logits
from adanet_amd.ops.optim import CosineLR, FusedAdam, FusedSGD, make_optimizer
from adanet_amd.ops.xent import softmax_xent

extension_available = _extension.available

__all__ = [
    "HipDropout", "HipLayerNorm", "HipLinear", "gemm_nt", "transpose2d",
    "weighted_sum_logits", "CosineLR", "FusedAdam", "FusedSGD",
    "make_optimizer", "softmax_xent", "extension_available",
]
