"""Distributed SyncBatchNorm (reference: /root/reference/module/sync_bn.py).

Forward all-reduces sum(x) and sum(x^2) ([F] vectors) and normalizes by the
GLOBAL train-node count `whole_size`; backward all-reduces dbias/dweight and
computes dx = (w/n)/std * (n*g - dbias - x_hat*dweight). Collectives run on
the main-thread process group (RCCL on GPU, gloo on CPU); the column
reductions themselves are [N,F]->[F] torch ops (rocBLAS/eager — profiled as
negligible next to SpMM).

Divergences from the reference implementation (the forward/backward algebra
itself must match numerically for parity, and is textbook sync-BN):
 - statistics are accumulated in fp32 even under bf16 compute (the reference
   is fp32-only);
 - the column sums go through our native two-phase colsum kernel on GPU
   (kernels.hip) instead of torch.sum — the thin-N eager reduction was an
   18.7 ms pathology (profiles/README.md);
 - explicit dtype casts at the normalize/return boundary so the module is
   usable inside a bf16 model.
"""
import torch
import torch.distributed as dist
from torch import nn
from torch.autograd import Function


def _colsum(x):
    from pipegcn_amd import native

    return native().colsum(x.contiguous())


def _maybe_all_reduce(t):
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)


class SyncBatchNormFunc(Function):
    @staticmethod
    def forward(ctx, x, weight, bias, whole_size, running_mean, running_var,
                training, momentum, eps):
        if not training:
            mean, var = running_mean, running_var
        else:
            # statistics in fp32 (bf16 products/sums lose too much precision)
            xs = x.float() if x.dtype != torch.float32 else x
            sum_x = _colsum(xs)
            sum_x2 = _colsum(xs * xs)
            _maybe_all_reduce(sum_x)
            _maybe_all_reduce(sum_x2)
            mean = sum_x / whole_size
            var = (sum_x2 - mean * sum_x) / whole_size
            running_mean.mul_(1 - momentum).add_(mean * momentum)
            running_var.mul_(1 - momentum).add_(var * momentum)
        std = torch.sqrt(var + eps)
        x_hat = (x - mean) / std  # fp32 (mean/std are fp32)
        if training:
            ctx.save_for_backward(x_hat, weight, std)
            ctx.whole_size = whole_size
        return (x_hat * weight + bias).to(x.dtype)

    @staticmethod
    def backward(ctx, grad):
        x_hat, weight, std = ctx.saved_tensors
        gs = grad.float() if grad.dtype != torch.float32 else grad
        dbias = _colsum(gs)
        dweight = _colsum(gs * x_hat.float())
        _maybe_all_reduce(dbias)
        _maybe_all_reduce(dweight)
        n = ctx.whole_size
        dx = (weight / n) / std * (n * grad - dbias - x_hat * dweight)
        return (dx.to(grad.dtype), dweight.to(weight.dtype),
                dbias.to(weight.dtype), None, None, None, None, None, None)


class SyncBatchNorm(nn.Module):
    def __init__(self, num_features, whole_size, eps=1e-5, momentum=0.1):
        super().__init__()
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.whole_size = whole_size
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))

    def forward(self, x):
        return SyncBatchNormFunc.apply(x, self.weight, self.bias,
                                       self.whole_size, self.running_mean,
                                       self.running_var, self.training,
                                       self.momentum, self.eps)
