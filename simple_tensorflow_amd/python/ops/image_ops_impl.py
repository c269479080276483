"""tf.image (reference python/ops/image_ops_impl.py subset: resize family,
flips, crops, standardization; resize kernels in csrc/kernels/cpu_image.cc)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import (RegisterGradient,
                                                        apply_op,
                                                        convert_to_tensor)
from simple_tensorflow_amd.python.ops import array_ops, math_ops, random_ops


class ResizeMethod(object):
    BILINEAR = 0
    NEAREST_NEIGHBOR = 1
    BICUBIC = 2
    AREA = 3


def resize_bilinear(images, size, align_corners=False, name=None):
    images = convert_to_tensor(images)
    size_t = convert_to_tensor(size, dtype=dtypes.int32)
    out = apply_op('ResizeBilinear', images, size_t,
                   align_corners=align_corners, name=name)
    if images._shape is not None and not isinstance(size, ops.Tensor):
        out.set_shape([images._shape[0], size[0], size[1],
                       images._shape[3]])
    return out


def resize_nearest_neighbor(images, size, align_corners=False, name=None):
    images = convert_to_tensor(images)
    size_t = convert_to_tensor(size, dtype=dtypes.int32)
    out = apply_op('ResizeNearestNeighbor', images, size_t,
                   align_corners=align_corners, name=name)
    if images._shape is not None and not isinstance(size, ops.Tensor):
        out.set_shape([images._shape[0], size[0], size[1],
                       images._shape[3]])
    return out


def resize_images(images, size, method=ResizeMethod.BILINEAR,
                  align_corners=False):
    if method == ResizeMethod.BILINEAR:
        return resize_bilinear(images, size, align_corners)
    if method == ResizeMethod.NEAREST_NEIGHBOR:
        return resize_nearest_neighbor(images, size, align_corners)
    raise NotImplementedError('resize method %r' % method)


@RegisterGradient('ResizeBilinear')
def _resize_bilinear_grad(op, grad):
    dx = apply_op('ResizeBilinearGrad', grad, op.inputs[0],
                  align_corners=op.get_attr('align_corners'))
    dx.set_shape(op.inputs[0]._shape)
    return [dx, None]


def flip_left_right(image):
    image = convert_to_tensor(image)
    return _reverse(image, axis=1)


def flip_up_down(image):
    image = convert_to_tensor(image)
    return _reverse(image, axis=0)


def _reverse(image, axis):
    # gather-based reverse along `axis` of a 3-D [h, w, c] image
    n = image._shape[axis]
    idx = ops.constant(list(range(n - 1, -1, -1)), dtype=dtypes.int32)
    if axis == 0:
        return array_ops.gather(image, idx)
    perm = [1, 0, 2]
    return array_ops.transpose(
        array_ops.gather(array_ops.transpose(image, perm), idx), perm)


def random_flip_left_right(image, seed=None):
    from simple_tensorflow_amd.python.ops import control_flow_ops
    u = random_ops.random_uniform([], 0.0, 1.0, seed=seed)
    return control_flow_ops.cond(math_ops.less(u, 0.5),
                                 lambda: flip_left_right(image),
                                 lambda: array_ops.identity(image))


def crop_to_bounding_box(image, offset_height, offset_width, target_height,
                         target_width):
    image = convert_to_tensor(image)
    if len(image._shape) == 3:
        return image[offset_height:offset_height + target_height,
                     offset_width:offset_width + target_width, :]
    return image[:, offset_height:offset_height + target_height,
                 offset_width:offset_width + target_width, :]


def central_crop(image, central_fraction):
    image = convert_to_tensor(image)
    h, w = image._shape[0], image._shape[1]
    ch = int(h * central_fraction)
    cw = int(w * central_fraction)
    oh = (h - ch) // 2
    ow = (w - cw) // 2
    return crop_to_bounding_box(image, oh, ow, ch, cw)


def random_crop(image, size, seed=None):
    image = convert_to_tensor(image)
    h, w = image._shape[0], image._shape[1]
    th, tw = size[0], size[1]
    oh = random_ops.random_uniform([], 0, h - th + 1, dtype=dtypes.int32,
                                   seed=seed) if h > th else 0
    ow = random_ops.random_uniform([], 0, w - tw + 1, dtype=dtypes.int32,
                                   seed=seed) if w > tw else 0
    begin = array_ops.stack([oh, ow, ops.constant(0)]) \
        if isinstance(oh, ops.Tensor) else [oh, ow, 0]
    out = array_ops.slice(image, begin, [th, tw, image._shape[2]])
    out.set_shape([th, tw, image._shape[2]])
    return out


def per_image_standardization(image):
    image = math_ops.cast(convert_to_tensor(image), dtypes.float32)
    num = 1
    for d in image._shape:
        num *= d
    mean = math_ops.reduce_mean(image)
    variance = math_ops.reduce_mean(math_ops.square(image)) - \
        math_ops.square(mean)
    stddev = math_ops.sqrt(math_ops.maximum(variance, ops.constant(0.0)))
    min_stddev = ops.constant(1.0 / float(num) ** 0.5)
    adj = math_ops.maximum(stddev, min_stddev)
    return (image - mean) / adj


def convert_image_dtype(image, dtype, saturate=False, name=None):
    image = convert_to_tensor(image)
    dtype = dtypes.as_dtype(dtype)
    if image.dtype == dtype:
        return image
    if image.dtype in (dtypes.uint8,) and dtype == dtypes.float32:
        return math_ops.cast(image, dtype) * ops.constant(1.0 / 255.0)
    if image.dtype == dtypes.float32 and dtype == dtypes.uint8:
        return math_ops.cast(image * ops.constant(255.0), dtype)
    return math_ops.cast(image, dtype)


def encode_png(image, compression=-1, name=None):
    """PNG encode via the python codec (py_func bridge)."""
    from simple_tensorflow_amd.python.lib.io import png_codec
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _enc(arr):
        return png_codec.encode_png(np.asarray(arr, dtype=np.uint8))

    return script_ops.py_func(_enc, [convert_to_tensor(image)],
                              dtypes.string, name=name)


def decode_png(contents, channels=0, dtype=dtypes.uint8, name=None):
    from simple_tensorflow_amd.python.lib.io import png_codec
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _dec(blob):
        b = blob if isinstance(blob, bytes) else bytes(blob)
        return png_codec.decode_png(b)

    out = script_ops.py_func(_dec, [convert_to_tensor(contents)],
                             dtypes.uint8, name=name)
    return out


def encode_jpeg(image, quality=95, name=None, **kw):
    """Baseline JFIF encode via the python codec
    (lib/io/jpeg_codec.py; reference EncodeJpeg / jpeg_mem.cc analog)."""
    from simple_tensorflow_amd.python.lib.io import jpeg_codec
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _enc(arr):
        return jpeg_codec.encode_jpeg(np.asarray(arr, dtype=np.uint8),
                                      quality=quality)

    return script_ops.py_func(_enc, [convert_to_tensor(image)],
                              dtypes.string, name=name)


def decode_jpeg(contents, channels=0, name=None, **kw):
    from simple_tensorflow_amd.python.lib.io import jpeg_codec
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _dec(blob):
        b = blob if isinstance(blob, bytes) else bytes(blob)
        img = jpeg_codec.decode_jpeg(b)
        if channels == 1 and img.shape[2] == 3:
            y = (0.299 * img[..., 0] + 0.587 * img[..., 1] +
                 0.114 * img[..., 2])
            img = np.clip(y, 0, 255).astype(np.uint8)[:, :, None]
        elif channels == 3 and img.shape[2] == 1:
            img = np.repeat(img, 3, 2)
        return img

    return script_ops.py_func(_dec, [convert_to_tensor(contents)],
                              dtypes.uint8, name=name)


def decode_gif(contents, name=None):
    """GIF decode to uint8 [num_frames, h, w, 3] via the python codec
    (lib/io/gif_codec.py; reference DecodeGif / gif_io.cc analog)."""
    from simple_tensorflow_amd.python.lib.io import gif_codec
    from simple_tensorflow_amd.python.ops import script_ops

    def _dec(blob):
        b = blob if isinstance(blob, bytes) else bytes(blob)
        return gif_codec.decode_gif(b)

    return script_ops.py_func(_dec, [convert_to_tensor(contents)],
                              dtypes.uint8, name=name)


def encode_gif(images, name=None):
    from simple_tensorflow_amd.python.lib.io import gif_codec
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _enc(arr):
        return gif_codec.encode_gif(np.asarray(arr, dtype=np.uint8))

    return script_ops.py_func(_enc, [convert_to_tensor(images)],
                              dtypes.string, name=name)


def decode_image(contents, channels=None, name=None):
    """Dispatches on magic bytes to decode_png / decode_jpeg / decode_gif
    (reference decode_image). GIF yields [frames, h, w, 3] like the
    reference."""
    from simple_tensorflow_amd.python.lib.io import (gif_codec, jpeg_codec,
                                                     png_codec)
    from simple_tensorflow_amd.python.ops import script_ops
    import numpy as np

    def _dec(blob):
        b = blob if isinstance(blob, bytes) else bytes(blob)
        if b[:2] == b'\xff\xd8':
            return jpeg_codec.decode_jpeg(b)
        if b[:4] == b'GIF8':
            return gif_codec.decode_gif(b)
        return png_codec.decode_png(b)

    return script_ops.py_func(_dec, [convert_to_tensor(contents)],
                              dtypes.uint8, name=name)


# ---------------------------------------------------------------------------
# color / photometric ops (reference image_ops_impl.py)
# ---------------------------------------------------------------------------
def rgb_to_hsv(images, name=None):
    return apply_op('RGBToHSV', convert_to_tensor(images,
                                                  dtype=dtypes.float32),
                    name=name)


def hsv_to_rgb(images, name=None):
    return apply_op('HSVToRGB', convert_to_tensor(images,
                                                  dtype=dtypes.float32),
                    name=name)


def rgb_to_grayscale(images, name=None):
    images = convert_to_tensor(images, dtype=dtypes.float32)
    w = ops.constant([0.2989, 0.587, 0.114], dtypes.float32)
    gray = math_ops.reduce_sum(images * w, -1, keep_dims=True)
    return gray


def grayscale_to_rgb(images, name=None):
    from simple_tensorflow_amd.python.ops import array_ops as ao
    images = convert_to_tensor(images, dtype=dtypes.float32)
    return ao.concat([images, images, images], -1)


def adjust_brightness(image, delta):
    return convert_to_tensor(image, dtype=dtypes.float32) + \
        convert_to_tensor(float(delta), dtype=dtypes.float32)


def adjust_contrast(images, contrast_factor, name=None):
    return apply_op('AdjustContrastv2',
                    convert_to_tensor(images, dtype=dtypes.float32),
                    convert_to_tensor(float(contrast_factor),
                                      dtype=dtypes.float32), name=name)


def adjust_saturation(image, saturation_factor, name=None):
    from simple_tensorflow_amd.python.ops import array_ops as ao
    hsv = rgb_to_hsv(image)
    h = hsv[..., 0:1]
    s = hsv[..., 1:2] * convert_to_tensor(float(saturation_factor))
    s = math_ops.minimum(s, ops.constant(1.0))
    v = hsv[..., 2:3]
    return hsv_to_rgb(ao.concat([h, s, v], -1))


def adjust_hue(image, delta, name=None):
    from simple_tensorflow_amd.python.ops import array_ops as ao
    hsv = rgb_to_hsv(image)
    h = hsv[..., 0:1] + convert_to_tensor(float(delta))
    h = h - math_ops.floor(h)  # wrap to [0, 1)
    return hsv_to_rgb(ao.concat([h, hsv[..., 1:2], hsv[..., 2:3]], -1))


def random_brightness(image, max_delta, seed=None):
    from simple_tensorflow_amd.python.ops import random_ops
    delta = random_ops.random_uniform([], -max_delta, max_delta, seed=seed)
    return convert_to_tensor(image, dtype=dtypes.float32) + delta


def random_contrast(image, lower, upper, seed=None):
    from simple_tensorflow_amd.python.ops import random_ops
    factor = random_ops.random_uniform([], lower, upper, seed=seed)
    return apply_op('AdjustContrastv2',
                    convert_to_tensor(image, dtype=dtypes.float32), factor)


def pad_to_bounding_box(image, offset_height, offset_width, target_height,
                        target_width):
    from simple_tensorflow_amd.python.ops import array_ops as ao
    image = convert_to_tensor(image)
    h, w = int(image._shape[-3]), int(image._shape[-2])
    pads = [[offset_height, target_height - offset_height - h],
            [offset_width, target_width - offset_width - w], [0, 0]]
    if len(image._shape) == 4:
        pads = [[0, 0]] + pads
    return ao.pad(image, pads)


def resize_image_with_crop_or_pad(image, target_height, target_width):
    image = convert_to_tensor(image)
    h, w = int(image._shape[-3]), int(image._shape[-2])
    # crop first
    ch = min(h, target_height)
    cw = min(w, target_width)
    image = crop_to_bounding_box(image, (h - ch) // 2, (w - cw) // 2, ch, cw)
    if ch == target_height and cw == target_width:
        return image
    return pad_to_bounding_box(image, (target_height - ch) // 2,
                               (target_width - cw) // 2, target_height,
                               target_width)


def total_variation(images, name=None):
    images = convert_to_tensor(images, dtype=dtypes.float32)
    if len(images._shape) == 3:
        dh = images[1:, :, :] - images[:-1, :, :]
        dw = images[:, 1:, :] - images[:, :-1, :]
        return math_ops.reduce_sum(math_ops.abs(dh)) + \
            math_ops.reduce_sum(math_ops.abs(dw))
    dh = images[:, 1:, :, :] - images[:, :-1, :, :]
    dw = images[:, :, 1:, :] - images[:, :, :-1, :]
    axes = [1, 2, 3]
    return math_ops.reduce_sum(math_ops.abs(dh), axes) + \
        math_ops.reduce_sum(math_ops.abs(dw), axes)


def non_max_suppression(boxes, scores, max_output_size, iou_threshold=0.5,
                        name=None):
    return apply_op('NonMaxSuppressionV2', convert_to_tensor(boxes),
                    convert_to_tensor(scores),
                    convert_to_tensor(int(max_output_size),
                                      dtype=dtypes.int32),
                    convert_to_tensor(float(iou_threshold)), name=name)


def sample_distorted_bounding_box(image_size, bounding_boxes,
                                  min_object_covered=0.1,
                                  aspect_ratio_range=None, area_range=None,
                                  max_attempts=100,
                                  use_image_if_no_bounding_boxes=False,
                                  seed=None, seed2=None, name=None):
    return apply_op(
        'SampleDistortedBoundingBox',
        convert_to_tensor(image_size, dtype=dtypes.int32),
        convert_to_tensor(bounding_boxes, dtype=dtypes.float32),
        min_object_covered=min_object_covered,
        aspect_ratio_range=list(aspect_ratio_range or [0.75, 1.33]),
        area_range=list(area_range or [0.05, 1.0]),
        max_attempts=max_attempts,
        use_image_if_no_bounding_boxes=use_image_if_no_bounding_boxes,
        seed=seed or 0, seed2=seed2 or 0, name=name)
