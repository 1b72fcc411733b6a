"""paddle.dataset (reference: python/paddle/dataset/ -- legacy
download-backed dataset readers).  Download-gated here (no network
egress); use paddle.io.Dataset / paddle.vision.datasets with local files."""


def _gone(name):
    def f(*a, **kw):
        raise RuntimeError(
            f"paddle.dataset.{name}: legacy downloader needs network egress; "
            "use paddle.io.Dataset over local files")
    return f


common = type("common", (), {"download": staticmethod(_gone("common.download"))})
