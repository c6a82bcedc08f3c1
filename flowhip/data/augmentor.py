"""Photometric + spatial augmentation for dense and sparse flow datasets.

Behavioral parity with the reference `core/utils/augmentor.py`: identical
probabilities, sampling distributions and crop/flip/eraser logic. Fresh
implementation on numpy + PIL (the reference used cv2 + torchvision
ColorJitter, neither of which this image ships): bilinear resizes go through
PIL (same half-pixel convention as cv2.INTER_LINEAR), and the color jitter
reproduces torchvision's sampling (uniform factor ranges, random op order).
Interpolation rounding differs from cv2 at the last bit; distributions and
semantics are identical.
"""

import numpy as np
from PIL import Image


class _ColorJitter:
    """torchvision-compatible ColorJitter on uint8 HWC numpy arrays.

    Factors: brightness/contrast/saturation ~ U[max(0,1-a), 1+a],
    hue ~ U[-h, h] (fraction of the hue circle), applied in a random order —
    matching torchvision.transforms.ColorJitter's get_params.
    """

    def __init__(self, brightness, contrast, saturation, hue):
        self.brightness = brightness
        self.contrast = contrast
        self.saturation = saturation
        self.hue = hue

    @staticmethod
    def _blend(a, b, factor):
        return np.clip(a * factor + b * (1.0 - factor), 0, 255)

    def _apply_brightness(self, img, f):
        return self._blend(img, np.zeros_like(img), f)

    def _apply_contrast(self, img, f):
        # torchvision: mean over the grayscale image
        gray = img[..., 0] * 0.299 + img[..., 1] * 0.587 + img[..., 2] * 0.114
        mean = gray.mean()
        return self._blend(img, np.full_like(img, mean), f)

    def _apply_saturation(self, img, f):
        gray = (img[..., 0] * 0.299 + img[..., 1] * 0.587 + img[..., 2] * 0.114)
        return self._blend(img, gray[..., None], f)

    def _apply_hue(self, img, f):
        hsv = np.array(Image.fromarray(img.astype(np.uint8)).convert("HSV"),
                       dtype=np.int16)
        hsv[..., 0] = (hsv[..., 0] + int(round(f * 255))) % 256
        rgb = Image.fromarray(hsv.astype(np.uint8), "HSV").convert("RGB")
        return np.array(rgb, dtype=np.float32)

    def __call__(self, img):
        img = img.astype(np.float32)
        ops = []
        if self.brightness > 0:
            f = np.random.uniform(max(0, 1 - self.brightness), 1 + self.brightness)
            ops.append(lambda im, f=f: self._apply_brightness(im, f))
        if self.contrast > 0:
            f = np.random.uniform(max(0, 1 - self.contrast), 1 + self.contrast)
            ops.append(lambda im, f=f: self._apply_contrast(im, f))
        if self.saturation > 0:
            f = np.random.uniform(max(0, 1 - self.saturation), 1 + self.saturation)
            ops.append(lambda im, f=f: self._apply_saturation(im, f))
        if self.hue > 0:
            f = np.random.uniform(-self.hue, self.hue)
            ops.append(lambda im, f=f: self._apply_hue(im, f))
        for i in np.random.permutation(len(ops)):
            img = ops[i](img)
        return img.astype(np.uint8)


def _resize_img(img, fx, fy):
    """Bilinear resize of a HWC uint8 image by scale factors (cv2.resize
    convention: out size = round(in * f))."""
    h, w = img.shape[:2]
    ow, oh = int(round(w * fx)), int(round(h * fy))
    return np.array(Image.fromarray(img).resize((ow, oh), Image.BILINEAR))


def _resize_flow(flow, fx, fy):
    """Bilinear resize of a HxWx2 float32 flow field (values unscaled)."""
    h, w = flow.shape[:2]
    ow, oh = int(round(w * fx)), int(round(h * fy))
    ch = [np.array(Image.fromarray(flow[..., i], mode="F").resize((ow, oh), Image.BILINEAR))
          for i in range(flow.shape[2])]
    return np.stack(ch, axis=-1)


class FlowAugmentor:
    """Dense-GT augmentation (reference augmentor.py:13-118)."""

    def __init__(self, crop_size, min_scale=-0.2, max_scale=0.5, do_flip=True):
        self.crop_size = crop_size
        self.min_scale = min_scale
        self.max_scale = max_scale
        self.spatial_aug_prob = 0.8
        self.stretch_prob = 0.8
        self.max_stretch = 0.2

        self.do_flip = do_flip
        self.h_flip_prob = 0.5
        self.v_flip_prob = 0.1

        self.photo_aug = _ColorJitter(0.4, 0.4, 0.4, 0.5 / 3.14)
        self.asymmetric_color_aug_prob = 0.2
        self.eraser_aug_prob = 0.5

    def color_transform(self, img1, img2):
        if np.random.rand() < self.asymmetric_color_aug_prob:
            img1 = self.photo_aug(img1)
            img2 = self.photo_aug(img2)
        else:
            stack = self.photo_aug(np.concatenate([img1, img2], axis=0))
            img1, img2 = np.split(stack, 2, axis=0)
        return img1, img2

    def eraser_transform(self, img1, img2, bounds=(50, 100)):
        ht, wd = img1.shape[:2]
        if np.random.rand() < self.eraser_aug_prob:
            mean_color = np.mean(img2.reshape(-1, 3), axis=0)
            for _ in range(np.random.randint(1, 3)):
                x0 = np.random.randint(0, wd)
                y0 = np.random.randint(0, ht)
                dx = np.random.randint(bounds[0], bounds[1])
                dy = np.random.randint(bounds[0], bounds[1])
                img2[y0:y0 + dy, x0:x0 + dx, :] = mean_color
        return img1, img2

    def spatial_transform(self, img1, img2, flow):
        ht, wd = img1.shape[:2]
        min_scale = np.maximum((self.crop_size[0] + 8) / float(ht),
                               (self.crop_size[1] + 8) / float(wd))

        scale = 2 ** np.random.uniform(self.min_scale, self.max_scale)
        scale_x = scale_y = scale
        if np.random.rand() < self.stretch_prob:
            scale_x *= 2 ** np.random.uniform(-self.max_stretch, self.max_stretch)
            scale_y *= 2 ** np.random.uniform(-self.max_stretch, self.max_stretch)
        scale_x = np.clip(scale_x, min_scale, None)
        scale_y = np.clip(scale_y, min_scale, None)

        if np.random.rand() < self.spatial_aug_prob:
            img1 = _resize_img(img1, scale_x, scale_y)
            img2 = _resize_img(img2, scale_x, scale_y)
            flow = _resize_flow(flow, scale_x, scale_y)
            flow = flow * [scale_x, scale_y]

        if self.do_flip:
            if np.random.rand() < self.h_flip_prob:
                img1 = img1[:, ::-1]
                img2 = img2[:, ::-1]
                flow = flow[:, ::-1] * [-1.0, 1.0]
            if np.random.rand() < self.v_flip_prob:
                img1 = img1[::-1, :]
                img2 = img2[::-1, :]
                flow = flow[::-1, :] * [1.0, -1.0]

        y0 = np.random.randint(0, img1.shape[0] - self.crop_size[0])
        x0 = np.random.randint(0, img1.shape[1] - self.crop_size[1])

        img1 = img1[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        img2 = img2[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        flow = flow[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        return img1, img2, flow

    def __call__(self, img1, img2, flow):
        img1, img2 = self.color_transform(img1, img2)
        img1, img2 = self.eraser_transform(img1, img2)
        img1, img2, flow = self.spatial_transform(img1, img2, flow)
        return (np.ascontiguousarray(img1), np.ascontiguousarray(img2),
                np.ascontiguousarray(flow, dtype=np.float32))


class SparseFlowAugmentor:
    """Sparse-GT (valid-mask) augmentation (reference augmentor.py:120-244):
    symmetric-only color jitter, nearest-point re-rasterization of the sparse
    flow on resize, margin-biased crops, h-flip only."""

    def __init__(self, crop_size, min_scale=-0.2, max_scale=0.5, do_flip=False):
        self.crop_size = crop_size
        self.min_scale = min_scale
        self.max_scale = max_scale
        self.spatial_aug_prob = 0.8
        self.stretch_prob = 0.8
        self.max_stretch = 0.2

        self.do_flip = do_flip
        self.h_flip_prob = 0.5
        self.v_flip_prob = 0.1

        self.photo_aug = _ColorJitter(0.3, 0.3, 0.3, 0.3 / 3.14)
        self.asymmetric_color_aug_prob = 0.2
        self.eraser_aug_prob = 0.5

    def color_transform(self, img1, img2):
        stack = self.photo_aug(np.concatenate([img1, img2], axis=0))
        img1, img2 = np.split(stack, 2, axis=0)
        return img1, img2

    def eraser_transform(self, img1, img2):
        ht, wd = img1.shape[:2]
        if np.random.rand() < self.eraser_aug_prob:
            mean_color = np.mean(img2.reshape(-1, 3), axis=0)
            for _ in range(np.random.randint(1, 3)):
                x0 = np.random.randint(0, wd)
                y0 = np.random.randint(0, ht)
                dx = np.random.randint(50, 100)
                dy = np.random.randint(50, 100)
                img2[y0:y0 + dy, x0:x0 + dx, :] = mean_color
        return img1, img2

    def resize_sparse_flow_map(self, flow, valid, fx=1.0, fy=1.0):
        ht, wd = flow.shape[:2]
        coords = np.meshgrid(np.arange(wd), np.arange(ht))
        coords = np.stack(coords, axis=-1).reshape(-1, 2).astype(np.float32)

        flow = flow.reshape(-1, 2).astype(np.float32)
        valid = valid.reshape(-1).astype(np.float32)

        coords0 = coords[valid >= 1]
        flow0 = flow[valid >= 1]

        ht1 = int(round(ht * fy))
        wd1 = int(round(wd * fx))

        coords1 = coords0 * [fx, fy]
        flow1 = flow0 * [fx, fy]

        xx = np.round(coords1[:, 0]).astype(np.int32)
        yy = np.round(coords1[:, 1]).astype(np.int32)

        v = (xx > 0) & (xx < wd1) & (yy > 0) & (yy < ht1)
        xx, yy, flow1 = xx[v], yy[v], flow1[v]

        flow_img = np.zeros([ht1, wd1, 2], dtype=np.float32)
        valid_img = np.zeros([ht1, wd1], dtype=np.int32)
        flow_img[yy, xx] = flow1
        valid_img[yy, xx] = 1
        return flow_img, valid_img

    def spatial_transform(self, img1, img2, flow, valid):
        ht, wd = img1.shape[:2]
        min_scale = np.maximum((self.crop_size[0] + 1) / float(ht),
                               (self.crop_size[1] + 1) / float(wd))

        scale = 2 ** np.random.uniform(self.min_scale, self.max_scale)
        scale_x = np.clip(scale, min_scale, None)
        scale_y = np.clip(scale, min_scale, None)

        if np.random.rand() < self.spatial_aug_prob:
            img1 = _resize_img(img1, scale_x, scale_y)
            img2 = _resize_img(img2, scale_x, scale_y)
            flow, valid = self.resize_sparse_flow_map(flow, valid, fx=scale_x, fy=scale_y)

        if self.do_flip:
            if np.random.rand() < 0.5:
                img1 = img1[:, ::-1]
                img2 = img2[:, ::-1]
                flow = flow[:, ::-1] * [-1.0, 1.0]
                valid = valid[:, ::-1]

        margin_y = 20
        margin_x = 50

        y0 = np.random.randint(0, img1.shape[0] - self.crop_size[0] + margin_y)
        x0 = np.random.randint(-margin_x, img1.shape[1] - self.crop_size[1] + margin_x)

        y0 = np.clip(y0, 0, img1.shape[0] - self.crop_size[0])
        x0 = np.clip(x0, 0, img1.shape[1] - self.crop_size[1])

        img1 = img1[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        img2 = img2[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        flow = flow[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        valid = valid[y0:y0 + self.crop_size[0], x0:x0 + self.crop_size[1]]
        return img1, img2, flow, valid

    def __call__(self, img1, img2, flow, valid):
        img1, img2 = self.color_transform(img1, img2)
        img1, img2 = self.eraser_transform(img1, img2)
        img1, img2, flow, valid = self.spatial_transform(img1, img2, flow, valid)
        return (np.ascontiguousarray(img1), np.ascontiguousarray(img2),
                np.ascontiguousarray(flow, dtype=np.float32),
                np.ascontiguousarray(valid))
