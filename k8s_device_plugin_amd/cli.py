"""CLI entry points: device plugin and node labeller daemons.

Flag surface matches the reference binaries:
  device plugin: -pulse, -resource_naming_strategy
    (reference: cmd/k8s-device-plugin/main.go:107-112)
  labeller: one boolean flag per label kind + DS_NODE_NAME env
    (reference: cmd/k8s-node-labeller/main.go:407-409,440)
Exit codes: 2 when the amdgpu/kfd driver is unavailable (reference:
amdgpu.go:157-160).
"""

from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading
import time

from . import __version__
from .topology import SysPaths, discover_gpus, DriverUnavailableError


def _setup_logging(verbose: int) -> None:
    logging.basicConfig(
        level=logging.DEBUG if verbose else logging.INFO,
        format="%(asctime)s %(levelname).1s %(name)s: %(message)s",
    )


def device_plugin_main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        prog="amd-device-plugin",
        description=f"AMD GPU device plugin for Kubernetes (MI355X-native) v{__version__}",
    )
    ap.add_argument("-pulse", "--pulse", type=int, default=0,
                    help="seconds between health checks (0 disables)")
    ap.add_argument("-resource_naming_strategy", "--resource_naming_strategy",
                    default="single", help="single or mixed")
    ap.add_argument("--kubelet-dir", default=None,
                    help="device-plugin dir override (tests)")
    ap.add_argument("--sysroot", default="/", help="sysfs root override (tests)")
    ap.add_argument("--server", default="native", choices=["native", "python"],
                    help="serving implementation (native C++/nghttp2 with "
                         "automatic python fallback, or python grpc)")
    ap.add_argument("--cdi", action="store_true",
                    help="write a CDI spec and return CDI device names from "
                         "Allocate (containerd>=1.7 / CRI-O with CDI)")
    ap.add_argument("--cdi-dir", default=None,
                    help="CDI spec directory (default /var/run/cdi)")
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="Prometheus metrics port (0 disables, the default)")
    ap.add_argument("--prestart-probe", action="store_true",
                    help="advertise pre_start_required and verify each "
                         "requested device answers before container start")
    ap.add_argument("--prestart-deep", action="store_true",
                    help="with --prestart-probe: additionally run the "
                         "MFMA/LDS/HBM deep probe (with performance "
                         "floors) on each requested GPU before start")
    ap.add_argument("--deep-probe-every", type=int, default=0,
                    help="run the deep GPU probe every Nth heartbeat and "
                         "pin floor-failing GPUs Unhealthy (0 = off; "
                         "floors via AMDXDP_MFMA_FLOOR_TFLOPS / "
                         "AMDXDP_HBM_FLOOR_GBPS)")
    ap.add_argument("--dump", action="store_true",
                    help="print discovered devices + allocator state as "
                         "JSON and exit (debugging)")
    ap.add_argument("--dump-podresources", action="store_true",
                    help="query the kubelet PodResources API and print "
                         "which pods hold amd.com/* devices, then exit")
    ap.add_argument("--podresources-socket",
                    default=None,
                    help="kubelet pod-resources socket (default "
                         "/var/lib/kubelet/pod-resources/kubelet.sock)")
    ap.add_argument("--exit-on-stream-loss", action="store_true",
                    help="exit(1) when a ListAndWatch stream breaks so the "
                         "DaemonSet restarts the pod (the ROCm plugin's "
                         "behavior); default is in-process re-registration")
    ap.add_argument("-v", "--verbose", action="count", default=0)
    args = ap.parse_args(argv)
    _setup_logging(args.verbose)
    log = logging.getLogger("amd-device-plugin")
    # startup banner (reference: main.go:94-121 logs name+version lines)
    log.info("AMD GPU device plugin for Kubernetes (MI355X-native) v%s",
             __version__)

    from .health import HeartbeatTicker
    from .plugin import (
        AMDGPUPlugin,
        PluginManager,
        StrategyError,
        get_resource_list,
        parse_strategy,
    )
    from .protos import deviceplugin as dp

    try:
        strategy = parse_strategy(args.resource_naming_strategy)
    except StrategyError as e:
        log.error("%s", e)
        return 1

    paths = SysPaths(args.sysroot)

    if args.dump_podresources:
        import json

        from .plugin.podresources import gpu_allocation_summary
        from .protos.podresources import PODRESOURCES_SOCKET

        sock = args.podresources_socket or PODRESOURCES_SOCKET
        try:
            print(json.dumps(gpu_allocation_summary(sock), indent=2))
        except Exception as e:
            log.error("podresources query failed: %s", e)
            return 1
        return 0

    # gate on the ROCm driver being present (reference: main.go:139-152)
    deadline = time.monotonic() + 60
    while not os.path.isdir(paths.kfd_class):
        if time.monotonic() > deadline:
            log.error("/sys/class/kfd not present; is the amdgpu driver loaded? (exit 2)")
            return 2
        log.warning("waiting for %s ...", paths.kfd_class)
        time.sleep(2)

    try:
        devices = discover_gpus(paths)
    except DriverUnavailableError as e:
        log.error("%s (exit 2)", e)
        return 2

    if args.dump:
        import json

        from .allocator import AllocationError, BestEffortPolicy
        from .topology import KFDTopology

        topo = KFDTopology.load(paths)
        out = {
            "devices": {i: d.as_dict() for i, d in devices.items()},
            "homogeneous": None,
            "allocator": None,
        }
        from .topology.discovery import is_homogeneous, unique_partition_config_count

        out["homogeneous"] = is_homogeneous(devices)
        out["partition_configs"] = unique_partition_config_count(devices)
        policy = BestEffortPolicy()
        try:
            policy.init([d for d in devices.values() if d.kfd_backed],
                        topology=topo)
            groups, node_of_id, weights = policy.export_state()
            out["allocator"] = {
                "groups": [{"parent": g[0], "node_ids": g[1]} for g in groups],
                "pair_weights": [
                    {"from": a, "to": b, "weight": w} for a, b, w in weights
                ],
                # closed-form fast path active? (uniform group-pair weights)
                "uniform_fast_path": policy._uniform,
            }
        except AllocationError as e:
            out["allocator"] = {"error": str(e)}
        print(json.dumps(out, indent=2))
        return 0

    try:
        resources = get_resource_list(devices, strategy)
    except StrategyError as e:
        log.error("%s", e)
        return 1
    if not resources:
        log.error("no AMD GPUs found; nothing to advertise")
        return 0
    log.info("advertising resources: %s", [f"amd.com/{r}" for r in resources])

    if args.cdi:
        from .plugin.cdi import CDI_SPEC_DIR, write_cdi_spec

        spec_path = write_cdi_spec(
            devices.values(), spec_dir=args.cdi_dir or CDI_SPEC_DIR
        )
        log.info("wrote CDI spec %s", spec_path)

    mgr = PluginManager(
        lambda res: AMDGPUPlugin(resource=res, paths=paths,
                                 cdi_enabled=args.cdi,
                                 cdi_spec_dir=args.cdi_dir,
                                 prestart_probe=args.prestart_probe,
                                 prestart_deep=args.prestart_deep,
                                 deep_probe_every=args.deep_probe_every,
                                 exit_on_stream_loss=args.exit_on_stream_loss),
        device_plugin_path=args.kubelet_dir or dp.DEVICE_PLUGIN_PATH,
        server_impl=args.server,
    )
    ticker = HeartbeatTicker(args.pulse)
    ticker.subscribe(mgr.heartbeat_all)

    if args.metrics_port:
        from .plugin.metrics import start_metrics_server

        start_metrics_server(mgr, args.metrics_port)

    stop = threading.Event()
    for sig in (signal.SIGINT, signal.SIGTERM, signal.SIGQUIT):
        signal.signal(sig, lambda *a: stop.set())

    mgr.run(resources)
    ticker.start()
    try:
        while not stop.wait(1.0):
            pass
    finally:
        log.info("shutting down")
        ticker.stop()
        mgr.stop()
    return 0


def labeller_main(argv=None) -> int:
    from .labeller import K8sClient, NodeLabelController, generate_labels
    from .labeller.labels import LABEL_KINDS

    ap = argparse.ArgumentParser(
        prog="amd-node-labeller",
        description=f"AMD GPU node labeller for Kubernetes (MI355X-native) v{__version__}",
    )
    for kind in LABEL_KINDS:
        flag = kind.replace("-", "_")
        ap.add_argument(f"-{flag}", f"--{flag}", action="store_true",
                        help=f"label nodes with {kind} properties")
    ap.add_argument("--sysroot", default="/", help="sysfs root override (tests)")
    ap.add_argument("--api-server", default=None,
                    help="k8s API base URL override (tests)")
    ap.add_argument("--oneshot", action="store_true",
                    help="reconcile once and exit (no watch)")
    ap.add_argument("--refresh-interval", type=int, default=0,
                    help="recompute labels every N seconds and reconcile on "
                         "change (0 = compute once at startup, like the "
                         "ROCm labeller)")
    ap.add_argument("-v", "--verbose", action="count", default=0)
    args = ap.parse_args(argv)
    _setup_logging(args.verbose)
    log = logging.getLogger("amd-node-labeller")

    node_name = os.environ.get("DS_NODE_NAME") or os.environ.get("NODE_NAME")
    if not node_name:
        log.error("DS_NODE_NAME env var not set")
        return 1

    enabled = {kind: getattr(args, kind.replace("-", "_")) for kind in LABEL_KINDS}
    if not any(enabled.values()):
        log.warning("no label kinds enabled; pass e.g. --vram --family --cu_count")

    paths = SysPaths(args.sysroot)
    labels = generate_labels(enabled, paths)
    log.info("computed %d label(s)", len(labels))

    client = K8sClient(base_url=args.api_server)
    ctl = NodeLabelController(client, node_name, labels)

    stop = threading.Event()
    for sig in (signal.SIGINT, signal.SIGTERM):
        signal.signal(sig, lambda *a: (stop.set(), ctl.stop()))

    if args.oneshot:
        ctl.reconcile()
        return 0

    if args.refresh_interval > 0:
        def refresher():
            while not stop.wait(args.refresh_interval):
                try:
                    fresh = generate_labels(enabled, paths)
                    patch = ctl.update_labels(fresh)
                    if patch:
                        log.info("labels changed; applied %d update(s)",
                                 len(patch))
                except Exception:
                    log.exception("label refresh failed")

        threading.Thread(target=refresher, daemon=True,
                         name="label-refresh").start()

    ctl.run(block=True)
    return 0


def partition_main(argv=None) -> int:
    """Admin tool: show or set SPX/CPX + NPS partition modes.

    Goes beyond the read-only reference (amdgpu.go:306-339); writes are
    double-gated (--allow flag AND AMDXDP_ALLOW_REPARTITION=1 env) because
    repartitioning tears down every kfd consumer on the node.
    """
    ap = argparse.ArgumentParser(
        prog="amd-partitionctl",
        description="Show or set AMD GPU compute/memory partition modes",
    )
    ap.add_argument("--sysroot", default="/", help="sysfs root override (tests)")
    ap.add_argument("--compute", default=None,
                    help="target compute mode (SPX/DPX/TPX/QPX/CPX)")
    ap.add_argument("--memory", default=None,
                    help="target memory mode (NPS1/NPS2/NPS4/NPS8)")
    ap.add_argument("--gpus", default=None,
                    help="comma-separated PCI addresses (default: all)")
    ap.add_argument("--allow", action="store_true",
                    help="confirm the destructive write (also requires "
                         "AMDXDP_ALLOW_REPARTITION=1 in the environment; "
                         "NEVER on shared boxes)")
    ap.add_argument("--settle-timeout", type=float, default=30.0)
    ap.add_argument("-v", "--verbose", action="count", default=0)
    args = ap.parse_args(argv)
    _setup_logging(args.verbose)

    import json

    from .topology import (
        PartitionError,
        available_partition_modes,
        current_partition_modes,
        set_partition_mode,
    )

    paths = SysPaths(args.sysroot)
    if args.compute is None and args.memory is None:
        avail_c, avail_m = available_partition_modes(paths)
        print(json.dumps({
            "current": {k: {"compute": c, "memory": m}
                        for k, (c, m) in current_partition_modes(paths).items()},
            "available_compute": avail_c,
            "available_memory": avail_m,
        }, indent=2))
        return 0

    try:
        modes = set_partition_mode(
            paths,
            compute=args.compute,
            memory=args.memory,
            pci_addrs=args.gpus.split(",") if args.gpus else None,
            allow=args.allow,
            settle_timeout_s=args.settle_timeout,
        )
    except PartitionError as e:
        print(f"error: {e}", file=sys.stderr)
        return 1
    print(json.dumps({k: {"compute": c, "memory": m}
                      for k, (c, m) in modes.items()}, indent=2))
    return 0


if __name__ == "__main__":
    prog = os.path.basename(sys.argv[0])
    if "labeller" in prog or (len(sys.argv) > 1 and sys.argv[1] == "labeller"):
        sys.exit(labeller_main(sys.argv[2:] if sys.argv[1:2] == ["labeller"] else None))
    if "partition" in prog or (len(sys.argv) > 1 and sys.argv[1] == "partition"):
        sys.exit(partition_main(sys.argv[2:] if sys.argv[1:2] == ["partition"] else None))
    sys.exit(device_plugin_main())
