from . import dp  # noqa
