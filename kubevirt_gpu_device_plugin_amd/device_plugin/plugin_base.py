"""Common DevicePlugin server lifecycle.

Python/grpcio equivalent of the reference's per-resource gRPC server:
socket creation, readiness self-dial, kubelet registration, ListAndWatch
health streaming, socket-removal restart
(reference: generic_device_plugin.go:217-351,619-697).

Differences from the reference, by design:
  * health transitions go through a lock+condition-versioned device list
    instead of unbuffered channels, so ListAndWatch reconnects (kubelet
    restarts) always see current state and health producers never block;
  * one inotify watch on the /dev/vfio directory covers every group node
    (the reference adds one fsnotify watch per device node and cannot see
    re-creation of a removed node, generic_device_plugin.go:647-678);
  * after a kubelet restart the restarted server keeps the controller's
    stop signal (the reference's restarted server detaches from the
    global stop channel, generic_device_plugin.go:283-285).
"""

import logging
import os
import threading
import time
from concurrent import futures
from dataclasses import dataclass

import grpc

from .. import dpapi
from . import consts, inotify

log = logging.getLogger(__name__)


@dataclass
class PluginConfig:
    """Host paths, injectable for tests (takes the role of the
    reference's rebindable package globals, device_plugin.go:70-79)."""
    device_plugin_dir: str = dpapi.DEVICE_PLUGIN_PATH
    kubelet_socket: str = dpapi.KUBELET_SOCKET
    vfio_dir: str = consts.VFIO_DEVICE_PATH
    iommu_dev: str = consts.IOMMU_DEVICE_PATH
    pci_base: str = consts.PCI_DEVICES_PATH
    namespace: str = consts.DEVICE_NAMESPACE
    connect_timeout_s: float = consts.CONNECTION_TIMEOUT_S
    # period of the ground-truth health resync pass; it is the only
    # health source while the /dev/vfio watch cannot be established and
    # a belt-and-braces check against missed inotify events otherwise
    health_resync_s: float = consts.HEALTH_RESYNC_S
    # kubelet-restart re-registration backoff (doubles up to the cap;
    # retries forever while the daemon lives)
    restart_backoff_initial_s: float = consts.RESTART_BACKOFF_INITIAL_S
    restart_backoff_max_s: float = consts.RESTART_BACKOFF_MAX_S


class DevicePluginBase(dpapi.DevicePluginServicer):
    """One kubelet-facing gRPC server per resource type."""

    def __init__(self, device_name, devices, config=None):
        self.device_name = device_name
        self.config = config or PluginConfig()
        self.socket_path = os.path.join(
            self.config.device_plugin_dir,
            "kubevirt-%s.sock" % device_name)
        self._devs = list(devices)   # dpapi.Device messages (mutable)
        self._lock = threading.Condition()
        self._version = 0
        self._server = None
        self._stop = None            # controller-owned threading.Event
        self._term = threading.Event()
        self._watch_armed = threading.Event()
        self._health_thread = None
        # serializes start/stop/restart between the controller thread
        # and the health thread's kubelet-restart path
        self._lifecycle = threading.RLock()
        # count of threads currently blocked in stop() waiting for
        # _lifecycle: the unbounded restart retry loop aborts when one
        # appears, so a rescan retiring this resource (or any external
        # stop) is never blocked behind a kubelet outage
        self._stop_waiters = 0
        self._stop_waiters_lock = threading.Lock()

    # ---- lifecycle ------------------------------------------------------

    @property
    def resource_name(self):
        return "%s/%s" % (self.config.namespace, self.device_name)

    def start(self, stop_event):
        """Create the socket, serve, self-dial, register with kubelet and
        start health watching (reference: Start, generic_device_plugin.go:217-257)."""
        with self._lifecycle:
            self._start_locked(stop_event)

    def _start_locked(self, stop_event):
        if self._server is not None:
            raise RuntimeError("gRPC server already started")
        self._stop = stop_event
        self._term = threading.Event()
        self._cleanup_socket()

        server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=8),
            options=(("grpc.so_reuseport", 0),
                     # Allocate sits on the pod-admission critical path
                     ("grpc.optimization_target", "latency")))
        dpapi.add_device_plugin_servicer(self, server)
        server.add_insecure_port("unix:" + self.socket_path)
        server.start()
        self._server = server

        try:
            self._start_watch_and_register()
        except (grpc.RpcError, grpc.FutureTimeoutError, OSError) as e:
            # A failed start must not leak a live server: without this,
            # the socket stays bound and the health thread stays alive,
            # and a later rescan() would create a second server on the
            # same socket path while the zombie can independently
            # re-register with a stale registry.
            log.error("[%s] error starting plugin server: %s",
                      self.device_name, e)
            self.stop()
            raise
        log.info("%s device plugin server ready", self.device_name)

    def _start_watch_and_register(self):
        self._wait_for_ready()

        # Arm the health watcher BEFORE registering: the moment kubelet
        # knows the socket it may remove/recreate it, and a removal
        # before the watcher is armed would be missed (the reference
        # registers first, generic_device_plugin.go:246-252 — benign
        # there only because fsnotify setup races the same way).
        # The thread binds THIS generation's term/stop events at spawn:
        # reading self._term lazily would let a thread from the previous
        # generation survive a restart (its term was replaced before it
        # woke) and react to our own stop()'s socket removal with a
        # spurious second restart.
        self._watch_armed = threading.Event()
        self._health_thread = threading.Thread(
            target=self._health_loop_guard,
            args=(self._term, self._stop),
            name="health-%s" % self.device_name, daemon=True)
        self._health_thread.start()
        self._watch_armed.wait(timeout=2.0)

        self.register()

    def stop(self):
        with self._stop_waiters_lock:
            self._stop_waiters += 1
        try:
            with self._lifecycle:
                if self._server is None:
                    return
                self._term.set()
                with self._lock:
                    self._lock.notify_all()
                server, self._server = self._server, None
                server.stop(grace=None)
                self._cleanup_socket()
        finally:
            with self._stop_waiters_lock:
                self._stop_waiters -= 1

    def restart(self, generation_term=None):
        """Full re-handshake after a kubelet restart
        (reference: restart, generic_device_plugin.go:275-286).

        Registration is retried with backoff: a restarting kubelet
        removes the plugin socket *before* its Registration service is
        back up, so the first re-register attempt routinely races it.
        (The reference gives up after one failed attempt and the plugin
        stays dead until the daemon is bounced,
        generic_device_plugin.go:688-692.)
        """
        with self._lifecycle:
            if self._should_exit():
                # the daemon is shutting down concurrently — a restart
                # here would resurrect a server the controller just
                # stopped
                return
            if generation_term is not None and generation_term.is_set():
                # the caller belongs to an already-retired server
                # generation (a restart completed while it was waking) —
                # a second restart would tear down the live server
                return
            log.info("restarting %s device plugin server",
                     self.device_name)
            stop_event = self._stop
            self.stop()
            # Retry forever with capped exponential backoff: a resource
            # is never abandoned while the daemon lives — a kubelet
            # outage of any length ends with re-registration.  (The old
            # behavior gave up after ~60 s; the reference gives up after
            # ONE attempt, generic_device_plugin.go:688-692.)
            backoff = self.config.restart_backoff_initial_s
            attempt = 0

            def abort():
                # daemon shutting down, or another thread is blocked in
                # stop() for this plugin (rescan retiring the resource)
                # — retrying further would hold _lifecycle against it
                return ((stop_event is not None and stop_event.is_set())
                        or self._stop_waiters > 0)

            while True:
                if abort():
                    return
                attempt += 1
                try:
                    self.start(stop_event)
                    return
                except (grpc.RpcError, grpc.FutureTimeoutError,
                        OSError) as e:
                    log.warning(
                        "[%s] restart attempt %d failed (%s); kubelet "
                        "may still be coming up — retrying in %.1fs",
                        self.device_name, attempt, e, backoff)
                    self.stop()
                    deadline = time.monotonic() + backoff
                    while time.monotonic() < deadline:
                        if abort():
                            return
                        if stop_event is not None:
                            stop_event.wait(0.1)
                        else:
                            time.sleep(0.1)
                    backoff = min(backoff * 2,
                                  self.config.restart_backoff_max_s)

    def _cleanup_socket(self):
        try:
            os.remove(self.socket_path)
        except FileNotFoundError:
            pass

    def _wait_for_ready(self):
        ch = grpc.insecure_channel("unix:" + self.socket_path)
        try:
            grpc.channel_ready_future(ch).result(
                timeout=self.config.connect_timeout_s)
        finally:
            ch.close()

    def register(self):
        """One-shot Registration RPC into kubelet
        (reference: Register, generic_device_plugin.go:289-310)."""
        ch = grpc.insecure_channel("unix:" + self.config.kubelet_socket)
        try:
            grpc.channel_ready_future(ch).result(
                timeout=self.config.connect_timeout_s)
            stub = dpapi.RegistrationStub(ch)
            stub.Register(
                dpapi.RegisterRequest(
                    version=dpapi.VERSION,
                    endpoint=os.path.basename(self.socket_path),
                    resource_name=self.resource_name),
                timeout=self.config.connect_timeout_s)
        finally:
            ch.close()
        log.info("[%s] registered with kubelet for resource %s",
                 self.device_name, self.resource_name)

    # ---- device state ---------------------------------------------------

    def devices_snapshot(self):
        with self._lock:
            return [dpapi.Device.FromString(d.SerializeToString())
                    for d in self._devs]

    def update_devices(self, devices):
        """Replace the advertised device list in place (hotplug rescan).
        ListAndWatch streams the new full list; no socket churn or
        re-registration — kubelet handles inventory changes over the
        existing stream.  Devices that persist keep their current
        health (a rescan must not quietly resurrect an Unhealthy GPU)."""
        with self._lock:
            old_health = {d.ID: d.health for d in self._devs}
            self._devs = list(devices)
            for d in self._devs:
                if d.ID in old_health:
                    d.health = old_health[d.ID]
            self._version += 1
            self._lock.notify_all()
        log.info("[%s] device list updated: %d devices",
                 self.device_name, len(devices))

    def set_health(self, device_ids, health):
        """Flip health for the given advertised IDs and wake ListAndWatch
        streams (takes the role of the healthy/unhealthy channels,
        generic_device_plugin.go:326-343)."""
        if isinstance(device_ids, str):
            device_ids = [device_ids]
        wanted = set(device_ids)
        with self._lock:
            changed = False
            for d in self._devs:
                if d.ID in wanted and d.health != health:
                    d.health = health
                    changed = True
            if changed:
                self._version += 1
                self._lock.notify_all()

    # ---- DevicePlugin service ------------------------------------------

    def GetDevicePluginOptions(self, request, context):  # noqa: N802
        return dpapi.DevicePluginOptions(
            pre_start_required=False,
            get_preferred_allocation_available=True)

    def PreStartContainer(self, request, context):  # noqa: N802
        return dpapi.PreStartContainerResponse()

    def ListAndWatch(self, request, context):  # noqa: N802
        """Initial full list, then a fresh full list on every health
        transition (reference: generic_device_plugin.go:313-350).

        Responses are built under the lock but yielded outside it —
        a slow kubelet consumer must never block the health producers
        or other streams.
        """
        log.info("[%s] ListAndWatch: sending %d devices",
                 self.device_name, len(self._devs))
        last = None
        while context.is_active():
            with self._lock:
                while last == self._version and not self._should_exit():
                    self._lock.wait(timeout=0.5)
                    if not context.is_active():
                        return
                if self._should_exit():
                    return
                last = self._version
                resp = dpapi.ListAndWatchResponse(devices=self._devs)
            yield resp

    def _should_exit(self):
        return self._term.is_set() or (self._stop is not None
                                       and self._stop.is_set())

    # ---- health ---------------------------------------------------------

    def _health_loop_guard(self, term, stop):
        try:
            self._health_loop(term, stop)
        except Exception:
            log.exception("[%s] health loop failed", self.device_name)

    def _group_to_ids(self):
        """Map watched /dev/vfio node names to advertised device IDs.
        Overridden by subclasses that know the registry."""
        return {}

    def _resync_health(self, group_to_ids):
        """Ground-truth pass: health = does the group's vfio node
        exist right now."""
        for group, ids in group_to_ids.items():
            exists = os.path.exists(
                os.path.join(self.config.vfio_dir, group))
            self.set_health(ids, dpapi.HEALTHY if exists
                            else dpapi.UNHEALTHY)

    def _health_loop(self, term, stop):
        """inotify loop: vfio node create/remove → health flips; removal
        of our own socket → kubelet restarted → full server restart
        (reference: healthCheck, generic_device_plugin.go:619-697).

        ``term``/``stop`` are this server generation's lifetime events,
        bound at thread spawn (see start())."""
        def should_exit():
            return term.is_set() or (stop is not None and stop.is_set())

        sock_base = os.path.basename(self.socket_path)
        with inotify.Watcher() as w:
            w.add_watch(self.config.device_plugin_dir)
            vfio_watched = False
            try:
                w.add_watch(self.config.vfio_dir)
                vfio_watched = True
            except OSError as e:
                log.warning("[%s] cannot watch %s (%s); falling back to "
                            "periodic ground-truth health resync",
                            self.device_name, self.config.vfio_dir, e)
            self._watch_armed.set()
            last_resync = time.monotonic()
            while not should_exit():
                events = w.read_events(timeout_s=0.2)
                # Ground-truth pass: the ONLY health source while the
                # vfio dir is unwatchable (otherwise devices would stay
                # Healthy forever with zero checks), and a low-frequency
                # safety net against missed inotify events when watched.
                if time.monotonic() - last_resync \
                        >= self.config.health_resync_s:
                    last_resync = time.monotonic()
                    if not vfio_watched:
                        # the dir may exist by now (vfio module loaded
                        # after daemon start) — prefer event-driven
                        try:
                            w.add_watch(self.config.vfio_dir)
                            vfio_watched = True
                            log.info("[%s] /dev/vfio watch established",
                                     self.device_name)
                        except OSError:
                            pass
                    self._resync_health(self._group_to_ids())
                if not events:
                    continue
                # recomputed per batch: device lists change in place on
                # hotplug rescans (update_devices)
                group_to_ids = self._group_to_ids()
                if any(ev.mask & inotify.IN_Q_OVERFLOW
                       for ev in events):
                    # kernel dropped events (flip storm): re-derive
                    # health from live node state instead of trusting
                    # the partial stream
                    log.warning("[%s] inotify queue overflow; "
                                "resyncing health from /dev/vfio",
                                self.device_name)
                    self._resync_health(group_to_ids)
                for ev in events:
                    if should_exit():
                        return
                    path = w.path_of(ev.wd)
                    if (path == self.config.device_plugin_dir
                            and ev.name == sock_base
                            and ev.mask & (inotify.IN_DELETE
                                           | inotify.IN_MOVED_FROM)):
                        log.info("[%s] socket removed; kubelet likely "
                                 "restarted", self.device_name)
                        self.restart(generation_term=term)
                        return
                    if path == self.config.vfio_dir \
                            and ev.name in group_to_ids:
                        ids = group_to_ids[ev.name]
                        if ev.mask & (inotify.IN_CREATE
                                      | inotify.IN_MOVED_TO):
                            self.set_health(ids, dpapi.HEALTHY)
                        elif ev.mask & (inotify.IN_DELETE
                                        | inotify.IN_MOVED_FROM):
                            log.info("[%s] vfio node %s removed; marking "
                                     "%s unhealthy", self.device_name,
                                     ev.name, ids)
                            self.set_health(ids, dpapi.UNHEALTHY)
