"""Fused multi-tensor optimizer / EMA kernels (gfx950 HIP).

Each call launches ONE kernel over a flattened list of tensor chunks
(RMSpropTF elementwise chain, AdamW, EMA),
replacing the reference's ~6 eager CUDA kernels per parameter tensor
(reference rmsprop_tf.py:86-120, timm/utils.py:329-340).
"""


from .extension import load_extension

_MAX_CHUNK = 1 << 20  # elements per chunk entry (kernel grid-strides anyway)


def rmsprop_tf_multi_tensor(params, grads, square_avgs, momentum_buffers,
                            lr, alpha, eps, momentum, weight_decay,
                            decoupled_decay, lr_in_momentum):
    ext = load_extension()
    ext.rmsprop_tf_multi_tensor(
        params, grads, square_avgs,
        momentum_buffers if momentum_buffers is not None else [],
        float(lr), float(alpha), float(eps), float(momentum), float(weight_decay),
        bool(decoupled_decay), bool(lr_in_momentum))


def adamw_multi_tensor(params, grads, exp_avgs, exp_avg_sqs, step,
                       lr, beta1, beta2, eps, weight_decay):
    ext = load_extension()
    ext.adamw_multi_tensor(
        params, grads, exp_avgs, exp_avg_sqs, int(step),
        float(lr), float(beta1), float(beta2), float(eps), float(weight_decay))


def ema_multi_tensor(ema_params, model_params, decay):
    ext = load_extension()
    ext.ema_multi_tensor(ema_params, model_params, float(decay))
