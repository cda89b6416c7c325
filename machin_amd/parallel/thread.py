"""Exception-piping threads.

Parity target: reference ``machin/parallel/thread.py`` (:9-73).
"""
import threading
import traceback


class ThreadException(Exception):
    pass


class Thread(threading.Thread):
    """Thread whose exceptions surface in the parent via watch()."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._exception = None

    def run(self):
        try:
            super().run()
        except Exception as e:  # noqa: BLE001 - surfaced via watch()
            self._exception = (repr(e), traceback.format_exc())

    @property
    def exception(self):
        return self._exception

    def watch(self):
        if self._exception is not None:
            raise ThreadException(
                f"Exception in thread {self.name}: "
                f"{self._exception[0]}\n{self._exception[1]}"
            )
