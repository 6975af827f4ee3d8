"""TTL register thread (parity: reference utils/register.py:22-86).

Writes a key bound to a TTL lease, then refreshes the lease at TTL/3.
If refresh fails repeatedly the register stops — the key expires and the
pod drops out of the cluster (the reference's failure-detection mechanism,
SURVEY.md §5.3)."""
import threading

from ..utils.errors import EdlRegisterError, EdlStoreError
from ..utils.log import get_logger
from .tables import ETCD_TTL

log = get_logger("edl.register")


class Register:
    def __init__(self, client, key, value, ttl=ETCD_TTL, exclusive=False):
        """exclusive=True -> put-if-absent (leader-election style); raises
        EdlRegisterError if the key is already held by someone else."""
        self._client = client
        self._key = key
        self._value = value
        self._ttl = ttl
        self._exclusive = exclusive
        self._stop = threading.Event()
        self._stopped_on_error = threading.Event()
        self._lease = None
        self._thread = None

    def start(self):
        self._lease = self._client.grant(self._ttl)
        if self._exclusive:
            acquired, _cur = self._client.put_if_absent(self._key, self._value, self._lease)
            if not acquired:
                self._client.revoke(self._lease)
                raise EdlRegisterError("key %s already held" % self._key)
        else:
            self._client.put(self._key, self._value, self._lease)
        self._thread = threading.Thread(target=self._refresh, daemon=True, name="register")
        self._thread.start()
        return self

    def update(self, value):
        self._value = value
        self._client.put(self._key, value, self._lease)

    def _refresh(self):
        fails = 0
        while not self._stop.wait(self._ttl / 3.0):
            try:
                if self._client.keepalive(self._lease):
                    fails = 0
                    continue
                fails += 1
            except EdlStoreError:
                fails += 1
            if fails >= 2:
                log.warning("register %s lost its lease; stopping", self._key)
                self._stopped_on_error.set()
                return

    @property
    def failed(self):
        """True if the register lost its lease (pod should treat itself as
        evicted — reference launcher.py:210-218 checks this)."""
        return self._stopped_on_error.is_set()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
        try:
            if self._lease is not None:
                self._client.revoke(self._lease)
        except EdlStoreError:
            pass
