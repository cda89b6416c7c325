"""Exception-piping processes.

Parity target: reference ``machin/parallel/process.py`` (:11-80):
a ``multiprocessing.Process`` subclass that captures child exceptions
(with traceback text) through a pipe; ``watch()`` re-raises them in
the parent as :class:`ProcessException`.
"""
import multiprocessing as mp
import traceback


class ProcessException(Exception):
    pass


class Process(mp.Process):
    """Process whose child exceptions surface in the parent."""

    def __init__(self, *args, ctx=mp, **kwargs):
        self._ctx_recv, self._ctx_send = ctx.Pipe(duplex=False)
        base = ctx.Process if ctx is not mp else mp.Process
        base.__init__(self, *args, **kwargs)
        self._exception = None

    def run(self):
        try:
            super().run()
            self._ctx_send.send(None)
        except Exception as e:  # noqa: BLE001 - piped to parent
            self._ctx_send.send((repr(e), traceback.format_exc()))

    @property
    def exception(self):
        if self._exception is None and self._ctx_recv.poll():
            self._exception = self._ctx_recv.recv()
        return self._exception

    def watch(self):
        """Re-raise any child exception in the parent."""
        exc = self.exception
        if exc is not None:
            raise ProcessException(
                f"Exception in process {self.pid}: {exc[0]}\n{exc[1]}"
            )
