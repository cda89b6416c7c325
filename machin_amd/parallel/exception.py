"""Cross-process exception transport.

Parity target: reference ``machin/parallel/exception.py``:
``ExceptionWithTraceback`` — wraps an exception raised in a worker so
the original traceback text survives pickling; unpickling rebuilds
the exception with a :class:`RemoteTraceback` cause on the master.
"""
import traceback


class RemoteTraceback(Exception):
    """Carries a worker's formatted traceback to the master side."""

    def __init__(self, tb: str):
        self.tb = tb

    def __str__(self):
        return self.tb


def _rebuild_exc(exc: Exception, tb: str):
    exc.__cause__ = RemoteTraceback(tb)
    return exc


class ExceptionWithTraceback:
    """Pickle an exception together with its traceback text; the
    unpickled object is the original exception chained to a
    RemoteTraceback cause."""

    def __init__(self, exc: Exception, tb=None):
        if tb is None:
            tb = exc.__traceback__
        text = "".join(
            traceback.format_exception(type(exc), exc, tb)
        )
        self.exc = exc
        self.tb = f'\n"""\n{text}"""'

    def __reduce__(self):
        return _rebuild_exc, (self.exc, self.tb)
