"""Full-net engine-vs-oracle parity (SURVEY §8d parity-run recipe): one
Forward+Backward of a multi-layer net, comparing EVERY top blob and EVERY
param diff against the oracle's independent restatement composed layer by
layer, at the reference's 1e-4 relative fp32 bar
(test_convolution_layer.cpp EXPECT_NEAR class).  This pins the Net wiring
(top/bottom routing, loss-weight seeding of the backward, diff zeroing)
— the per-op golden tests in test_oracle.py can't see wiring bugs.
"""
import numpy as np

from engine_util import TOL, relerr, net_from_text
import caffe_amd as ca

from oracle import oracle  # noqa: E402  (engine_util puts REPO on sys.path)

NET = """name: "chain"
layer {
  name: "input"
  type: "Input"
  top: "in0"
  top: "in1"
  input_param {
    shape { dim: 4 dim: 3 dim: 8 dim: 8 }
    shape { dim: 4 }
  }
}
layer {
  name: "c1"
  type: "Convolution"
  bottom: "in0"
  top: "conv"
  convolution_param { num_output: 4 kernel_size: 3 pad: 1
    weight_filler { type: "gaussian" std: 0.3 }
    bias_filler { type: "gaussian" std: 0.1 } }
}
layer { name: "r1" type: "ReLU" bottom: "conv" top: "relu" }
layer {
  name: "p1"
  type: "Pooling"
  bottom: "relu"
  top: "pool"
  pooling_param { pool: MAX kernel_size: 2 stride: 2 }
}
layer {
  name: "ip1"
  type: "InnerProduct"
  bottom: "pool"
  top: "fc"
  inner_product_param { num_output: 5
    weight_filler { type: "gaussian" std: 0.2 }
    bias_filler { type: "constant" value: 0.05 } }
}
layer {
  name: "loss"
  type: "SoftmaxWithLoss"
  bottom: "fc"
  bottom: "in1"
  top: "out"
}
"""


def test_fullnet_forward_backward_matches_oracle():
    ca.set_mode("cpu")
    ca.set_random_seed(23)
    net = net_from_text(NET)
    rng = np.random.default_rng(5)
    x = rng.standard_normal((4, 3, 8, 8)).astype(np.float32)
    labels = np.array([1, 0, 4, 2], np.float32)
    net.set_blob("in0", x)
    net.set_blob("in1", labels)
    net.forward()
    net.backward()

    # engine params, by (layer, blob) — fillers already ran
    params = {(ln, bi): net.param(i)
              for i, (ln, bi, _) in
              ((i, net.param_info(i)) for i in range(net.num_params()))}
    cw = params[("c1", 0)].reshape(4, 3, 3, 3)
    cb = params[("c1", 1)]
    # IP weight is [Nout, K]
    iw = params[("ip1", 0)].reshape(5, -1)
    ib = params[("ip1", 1)]

    # oracle forward chain (float64 accumulation inside oracle ops)
    o_conv = oracle.conv_fwd(x, cw, cb, pad=(1, 1), stride=(1, 1))
    o_relu = oracle.relu_fwd(o_conv)
    o_pool, o_mask = oracle.pool_max_fwd(o_relu, 2, 2, 0, 0, 2, 2)
    o_fc = oracle.ip_fwd(o_pool.reshape(4, -1), iw, ib)
    o_prob = oracle.softmax_fwd(o_fc, 4, 5, 1)
    o_loss = oracle.softmaxloss_fwd(o_prob, labels, 4, 5, 1)

    assert relerr(net.blob("conv"), o_conv) < TOL
    assert relerr(net.blob("relu"), o_relu) < TOL
    assert relerr(net.blob("pool"), o_pool) < TOL
    assert relerr(net.blob("fc"), o_fc) < TOL
    assert abs(net.blob("out")[0] - o_loss) < TOL * max(1.0, abs(o_loss))

    # oracle backward chain (loss weight 1)
    d_fc = oracle.softmaxloss_bwd(o_prob, labels, 4, 5, 1)
    d_pool_flat, d_iw, d_ib = oracle.ip_bwd(
        o_pool.reshape(4, -1), iw, d_fc)
    d_relu = oracle.pool_max_bwd(
        d_pool_flat.reshape(o_pool.shape), o_mask,
        o_relu.shape[2], o_relu.shape[3])
    d_conv = oracle.relu_bwd(o_conv, d_relu)
    _, d_cw, d_cb = oracle.conv_bwd(x, cw, d_conv, pad=(1, 1),
                                    stride=(1, 1), want_db=True)

    diffs = {(ln, bi): net.param(i, diff=True)
             for i, (ln, bi, _) in
             ((i, net.param_info(i)) for i in range(net.num_params()))}
    assert relerr(diffs[("ip1", 0)], d_iw.ravel()) < TOL
    assert relerr(diffs[("ip1", 1)], d_ib.ravel()) < TOL
    assert relerr(diffs[("c1", 0)], d_cw.ravel()) < TOL
    assert relerr(diffs[("c1", 1)], d_cb.ravel()) < TOL
