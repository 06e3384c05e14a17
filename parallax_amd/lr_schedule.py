"""Learning-rate schedules from the reference CNN benchmark.

Reference: models/resnet_model.py:232-255 (5-epoch linear warmup from
0.1 to batch-scaled LR, then /10 at epochs 30/60/80),
models/resnet_model.py:326-331 and densenet_model.py:96-103 (cifar10
piecewise schedules), benchmark_cnn.py get_learning_rate (exponential
decay when --learning_rate/--num_epochs_per_decay are given).

Schedules are plain step -> float functions; drivers apply them with
``set_lr(optimizer, lr)`` before each step (the optimizers read
group['lr'] per launch in eager mode).  Not compatible with a captured
optimizer hipGraph (the capture freezes the scalar LR; bench.py's
captured paths run at fixed LR, matching the reference's constant-LR
benchmark setting).
"""

IMAGENET_NUM_TRAIN_IMAGES = 1281167
CIFAR10_NUM_TRAIN_IMAGES = 50000


def piecewise_constant(step, boundaries, values):
    """tf.train.piecewise_constant: values[i] for
    boundaries[i-1] < step <= boundaries[i] (len(values) =
    len(boundaries) + 1)."""
    for b, v in zip(boundaries, values):
        if step <= b:
            return v
    return values[-1]


def exponential_decay(initial_lr, step, decay_steps, decay_rate,
                      staircase=True):
    """tf.train.exponential_decay (benchmark_cnn.py custom-LR path)."""
    p = step / float(decay_steps)
    if staircase:
        p = int(p)
    return initial_lr * (decay_rate ** p)


def resnet_imagenet_lr(step, batch_size, num_workers=1):
    """resnet_model.py:232-255 — per-GPU batch_size; the epoch length
    uses the GLOBAL batch."""
    global_batch = batch_size * num_workers
    batches_per_epoch = IMAGENET_NUM_TRAIN_IMAGES / float(global_batch)
    warmup_batches = batches_per_epoch * 5
    lr0 = 0.1
    lr_after = global_batch / 256.0 * 0.1 if global_batch > 256 else 0.1
    if step <= int(batches_per_epoch * 5):
        return lr0 + (lr_after - lr0) / warmup_batches * step
    boundaries = [int(batches_per_epoch * x) for x in (5, 30, 60, 80)]
    values = [None] + [lr_after / 10 ** i for i in range(4)]
    return piecewise_constant(step, boundaries[1:], values[1:])


def resnet_cifar_lr(step, batch_size, num_workers=1):
    """resnet_model.py:326-331 — [82, 123] epoch boundaries."""
    bpe = int(CIFAR10_NUM_TRAIN_IMAGES / (batch_size * num_workers))
    return piecewise_constant(step, [bpe * 82, bpe * 123, bpe * 300],
                              [0.1, 0.01, 0.001, 0.0002])


def densenet_cifar_lr(step, batch_size, num_workers=1):
    """densenet_model.py:96-103 — [150, 225] epoch boundaries."""
    bpe = int(CIFAR10_NUM_TRAIN_IMAGES / (batch_size * num_workers))
    return piecewise_constant(step, [bpe * 150, bpe * 225, bpe * 300],
                              [0.1, 0.01, 0.001, 0.0001])


def model_lr_schedule(model_name, dataset='imagenet'):
    """Return the reference per-model schedule fn(step, batch_size,
    num_workers) -> lr, or None when the model has no published schedule
    (drivers then keep their fixed --learning_rate)."""
    if dataset == 'cifar10':
        if model_name.startswith('densenet'):
            return densenet_cifar_lr
        if model_name.startswith('resnet'):
            return resnet_cifar_lr
        return None
    if model_name.startswith('resnet'):
        return resnet_imagenet_lr
    return None


def set_lr(optimizer, lr):
    for group in optimizer.param_groups:
        group['lr'] = lr
