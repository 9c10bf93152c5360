"""NHWC layer modules over the native op layer."""

import math

import torch
import torch.nn as nn

from ..ops import functional as AF


class Conv2dNHWC(nn.Module):
    """Bias-free conv; weight (K, R, S, C) fp32 master."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.weight = nn.Parameter(
            torch.empty(out_channels, kernel_size, kernel_size, in_channels))
        self.reset_parameters()

    def reset_parameters(self):
        # kaiming normal, fan_out = K * R * S (matches torch's fan_out for a
        # (K,C,R,S) conv weight; reference init at models/utils.py:5-10)
        fan_out = self.out_channels * self.kernel_size * self.kernel_size
        with torch.no_grad():
            self.weight.normal_(0, math.sqrt(2.0 / fan_out))

    def forward(self, x):
        return AF.conv2d(x, self.weight, self.stride, self.padding)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, k={self.kernel_size}, "
                f"s={self.stride}, p={self.padding}, layout=KRSC")


class BatchNormAct2d(nn.Module):
    """BatchNorm over NHWC channels with fused optional residual-add + ReLU.

    ``sync_group`` (set by parallel.convert_sync_batchnorm) enables SyncBN
    stat exchange; ``self.training`` False -> frozen running stats with grads
    flowing (the reference's net.eval()-while-training mode,
    strategy.py:366-367).
    """

    def __init__(self, num_features, relu=True, eps=1e-5, momentum=0.1):
        super().__init__()
        self.num_features = num_features
        self.relu = relu
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.sync = False          # enabled by parallel.convert_sync_batchnorm
        self.sync_group = None     # None -> default process group when sync

    def _pg(self):
        if not self.sync:
            return None
        if self.sync_group is not None:
            return self.sync_group
        import torch.distributed as dist
        return dist.group.WORLD if dist.is_initialized() else None

    def forward(self, x, residual=None):
        return AF.batch_norm_act(x, self.weight, self.bias, self.running_mean,
                                 self.running_var, self.training, self.momentum,
                                 self.eps, self.relu, residual, self._pg(),
                                 fuse_backward=True)

    def extra_repr(self):
        return f"{self.num_features}, relu={self.relu}"

    def __getstate__(self):
        state = self.__dict__.copy()
        return state


class ConvTranspose2dNHWC(nn.Module):
    """Transposed conv (VAE decoder). Weight (K=C_in, R, S, C=C_out) fp32;
    forward is the conv bwd-data computation, so the same HIP kernels serve
    both (SURVEY.md §2.4 VAE row)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0,
                 bias=False):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.weight = nn.Parameter(
            torch.empty(in_channels, kernel_size, kernel_size, out_channels))
        self.bias = nn.Parameter(torch.zeros(out_channels)) if bias else None
        fan_out = out_channels * kernel_size * kernel_size
        with torch.no_grad():
            self.weight.normal_(0, math.sqrt(2.0 / fan_out))

    def forward(self, x):
        y = _TransposedConv2d.apply(x, self.weight, self.stride, self.padding)
        if self.bias is not None:
            y = y + self.bias.to(y.dtype)
        return y


class _TransposedConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride, padding):
        from ..ops.functional import cast_cached, wt_cached
        w_c = cast_cached(weight, x.dtype)
        ctx.save_for_backward(x)
        ctx.w_c = w_c
        ctx.stride, ctx.padding = stride, padding
        ctx.weight_dtype = weight.dtype
        k = w_c.shape[1]
        h_out = (x.shape[1] - 1) * stride - 2 * padding + k
        w_out = (x.shape[2] - 1) * stride - 2 * padding + k
        if x.is_cuda and x.dtype == torch.bfloat16:
            from ..ops.extension import require_extension
            # convT fwd == conv bwd-data; its "wt" is (Cout,R,S,Cin)
            return require_extension().conv2d_bwd_data(x, wt_cached(w_c), stride,
                                                       padding, h_out, w_out)
        xc = x.permute(0, 3, 1, 2).float()
        wc = w_c.permute(0, 3, 1, 2).float()  # (C_in, C_out, R, S)
        y = torch.nn.functional.conv_transpose2d(xc, wc, stride=stride, padding=padding)
        return y.to(x.dtype).permute(0, 2, 3, 1).contiguous()

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        w_c = ctx.w_c
        dy = dy.contiguous()
        dx = dw = None
        if x.is_cuda and x.dtype == torch.bfloat16:
            from ..ops.extension import require_extension
            ext = require_extension()
            if ctx.needs_input_grad[0]:
                dx = ext.conv2d_fwd(dy, w_c, ctx.stride, ctx.padding)
            if ctx.needs_input_grad[1]:
                # dW[cin, r, s, cout] = conv-bwd-weight with roles swapped
                dw = ext.conv2d_bwd_weight(x, dy, w_c.shape[1], w_c.shape[2],
                                           ctx.stride, ctx.padding)
        else:
            xc = x.permute(0, 3, 1, 2).float()
            wc = w_c.permute(0, 3, 1, 2).float()
            dyc = dy.permute(0, 3, 1, 2).float()
            if ctx.needs_input_grad[0]:
                dx = torch.nn.functional.conv2d(dyc, wc, stride=ctx.stride,
                                                padding=ctx.padding)
                dx = dx.to(x.dtype).permute(0, 2, 3, 1).contiguous()
            if ctx.needs_input_grad[1]:
                dw = torch.nn.grad.conv2d_weight(dyc, list(wc.shape), xc,
                                                 stride=ctx.stride, padding=ctx.padding)
                dw = dw.permute(0, 2, 3, 1).contiguous()
        if dw is not None:
            dw = dw.to(ctx.weight_dtype)
        return dx, dw, None, None


def nchw_to_nhwc(x):
    return x.permute(0, 2, 3, 1).contiguous()


def nhwc_to_nchw(x):
    return x.permute(0, 3, 1, 2).contiguous()
