"""The mnist_replica MLP: 784 -> hidden(100) relu -> 10 softmax-xent.

Same architecture/hyperparameters as the reference benchmark workload
(``examples/mnist/mnist_replica.py:122-145``: truncated-normal init with
stddev 1/sqrt(784), hidden 100, batch 100, lr 0.01). Forward AND backward
are written explicitly against ``tfmesos_amd.ops`` (bf16 MFMA GEMM +
fused softmax-xent + fused relu-bwd on GPU) — no autograd in the hot
path, so the step is a short fixed kernel sequence (5 kernels at the
benchmark geometry; GraphedStep can capture it and self-tunes
graph-vs-eager).
"""

import math

import torch

from tfmesos_amd import ops


class MnistMLP(object):

    def __init__(self, hidden_units=100, image_pixels=784, classes=10,
                 seed=1234):
        self.hidden = hidden_units
        self.inputs = image_pixels
        self.classes = classes
        self.seed = seed

    def param_specs(self):
        return [
            ("hid_w", (self.inputs, self.hidden)),
            ("hid_b", (self.hidden,)),
            ("sm_w", (self.hidden, self.classes)),
            ("sm_b", (self.classes,)),
        ]

    def init_params(self):
        """Truncated-normal(stddev=1/sqrt(inputs)) weights, zero biases —
        the reference's init (mnist_replica.py:122-137)."""
        g = torch.Generator().manual_seed(self.seed)
        std1 = 1.0 / math.sqrt(self.inputs)
        std2 = 1.0 / math.sqrt(self.hidden)

        def trunc(shape, std):
            t = torch.empty(*shape)
            torch.nn.init.trunc_normal_(t, std=std, a=-2 * std, b=2 * std,
                                        generator=g)
            return t

        return [
            ("hid_w", trunc((self.inputs, self.hidden), std1)),
            ("hid_b", torch.zeros(self.hidden)),
            ("sm_w", trunc((self.hidden, self.classes), std2)),
            ("sm_b", torch.zeros(self.classes)),
        ]

    def fwd_bwd(self, p, x, y, g):
        """One replica fwd+bwd.

        p: callable name -> bf16 (GPU) / fp32 (CPU) param view
        x: [B, 784] activations dtype; y: [B] int64 labels
        g: callable name -> grad view to fill (fp32)
        Returns mean loss (fp32 scalar tensor).
        """
        B = x.shape[0]
        hid_w, hid_b = p("hid_w"), p("hid_b")
        sm_w, sm_b = p("sm_w"), p("sm_b")

        # fwd hidden layer: split-K GEMM + bias + relu
        h = ops.gemm_bias_act(x, hid_w, hid_b, act="relu")       # [B,H]
        if x.is_cuda and B <= 128 and self.hidden <= 128 \
                and self.classes <= 16:
            # the whole classifier head in ONE MFMA kernel
            # (csrc/softmax_xent.hip): logits GEMM + softmax + loss +
            # dlogits + relu-masked dh + dW2/db2. At this size each
            # launch is ~4-7 us of execution floor, so kernel COUNT is
            # the step time: 5 launches total (fwd GEMM stripes, its
            # reduce, head, dW1, apply). Two measured dead ends kept
            # out: forking dW2 to a side stream (event edges cost more
            # than the GEMM they overlap), and consuming the fwd
            # split-K stripes directly in the head
            # (ops.mlp_fwd_head_fused — a SINGLE workgroup pulling
            # 360 KB of stripes from 8 XCDs' L2s took ~36 us; one-WG
            # consumers must read tiny inputs, so the many-WG reduce
            # kernel stays).
            loss, dlogits, dh = ops.mlp_head_fused(
                h, sm_w, sm_b, y, dw2=g("sm_w"), db2=g("sm_b"))
            ops.gemm_bias_act(x, dh, trans_a=True, out=g("hid_w"),
                              colsum_out=g("hid_b"))
            return loss
        logits = ops.gemm_bias_act(h, sm_w, sm_b, act="none")
        loss, dlogits = ops.softmax_xent_fused(logits, y)
        dh = ops.gemm_bias_act(dlogits, sm_w, trans_b=True,
                               act="relu_bwd", aux=h)
        # dW GEMMs with the bias-grad colsums fused in; fp32 grads
        # written straight into the flat grad views
        ops.gemm_bias_act(h, dlogits, trans_a=True, out=g("sm_w"),
                          colsum_out=g("sm_b"))
        ops.gemm_bias_act(x, dh, trans_a=True, out=g("hid_w"),
                          colsum_out=g("hid_b"))
        return loss

    def supports_fused_apply(self, trainer, x):
        """The single-kernel-tail fast path: world==1 colocated, plain
        SGD, GPU, and the head-kernel shape limits."""
        return (x.is_cuda and trainer.world == 1
                and trainer.store.opt == "sgd"
                and float(trainer.store.hparams.get("momentum", 0.0)) == 0.0
                and float(trainer.store.hparams.get("weight_decay", 0.0)) == 0.0
                and x.shape[0] <= 128 and self.hidden <= 128
                and self.classes <= 16)

    def fwd_bwd_apply(self, trainer, x, y, lr):
        """One fused replica step (world==1 fast path): fwd GEMM ->
        head (loss, dh, classifier grads) -> ONE tail kernel that
        computes dW1 and applies SGD to all four params in its
        epilogue (ops.mlp_tail_sgd). 4 kernels/step vs 5 — the flat
        sgd apply launch disappears; same math as fwd_bwd + step
        (plain SGD, grad mean over the single worker = 1).
        """
        st = trainer.store
        p = lambda n: st.view(n, bf16=True)
        g = trainer.grad_view
        h = ops.gemm_bias_act(x, p("hid_w"), p("hid_b"), act="relu")
        loss, _, dh = ops.mlp_head_fused(
            h, p("sm_w"), p("sm_b"), y, dw2=g("sm_w"), db2=g("sm_b"))
        ops.mlp_tail_sgd(
            x, dh, st.view("hid_w"), p("hid_w"), st.view("hid_b"),
            p("hid_b"), g("sm_w"), st.view("sm_w"), p("sm_w"),
            g("sm_b"), st.view("sm_b"), p("sm_b"), lr)
        st.global_step += 1
        return loss

    def loss_only(self, p, x, y):
        h = ops.gemm_bias_act(x, p("hid_w"), p("hid_b"), act="relu")
        logits = ops.gemm_bias_act(h, p("sm_w"), p("sm_b"))
        loss, _ = ops.softmax_xent_fwd(logits, y)
        return loss


def synthetic_batch(batch_size=100, image_pixels=784, classes=10,
                    device="cpu", dtype=torch.float32, seed=0):
    """Synthetic MNIST-shaped data (no network for datasets; BASELINE.md
    prescribes synthetic data / random-init weights)."""
    gen = torch.Generator().manual_seed(seed)
    x = torch.rand(batch_size, image_pixels, generator=gen)
    y = torch.randint(0, classes, (batch_size,), generator=gen)
    return x.to(device=device, dtype=dtype), y.to(device)
