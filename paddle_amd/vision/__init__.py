from . import datasets  # noqa: F401
from . import models  # noqa: F401
from . import transforms  # noqa: F401


# image backend plumbing (reference: vision/image.py)
_image_backend = "pil"


def set_image_backend(backend):
    global _image_backend
    assert backend in ("pil", "cv2", "tensor")
    _image_backend = backend


def get_image_backend():
    return _image_backend


def image_load(path, backend=None):
    backend = backend or _image_backend
    try:
        from PIL import Image
        return Image.open(path)
    except ImportError as e:
        raise RuntimeError("image_load needs Pillow (not in this image); "
                           "decode with numpy and use paddle.to_tensor") from e


from . import datasets  # noqa: F401

from . import ops  # noqa: F401
