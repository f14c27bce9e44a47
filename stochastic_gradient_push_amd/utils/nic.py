"""Network-interface autodiscovery for RCCL/Gloo env pinning.

Parity: reference experiment_utils/helpers.py:44-67.  RCCL honors the
same ``NCCL_SOCKET_IFNAME`` / ``NCCL_IB_DISABLE`` environment variables
as NCCL, so the ethernet-vs-infiniband switch carries over unchanged.
"""

import os
import subprocess

_PREFIXES = {
    "ethernet": ("ens", "eth", "enp"),
    "infiniband": ("ib",),
}


def get_tcp_interface_name(network_interface_type: str = "ethernet") -> str:
    """Return the name of an interface of the requested type that is up."""
    interfaces = os.listdir("/sys/class/net")
    out = subprocess.run(
        ["ip", "link", "show", "up"], capture_output=True, check=False
    ).stdout.decode("utf-8", errors="replace")

    prefixes = _PREFIXES[network_interface_type]
    for iface in interfaces:
        if iface.startswith(prefixes) and iface in out:
            print(f"Using network interface {iface}")
            return iface
    print("List of network interfaces found:", interfaces)
    print("Prefix list being used to search:", prefixes)
    raise RuntimeError(
        f"No usable {network_interface_type} interface found"
    )


def pin_comm_env(backend: str, network_interface_type: str) -> None:
    """Export the RCCL/Gloo socket-interface env vars for the chosen
    backend/fabric (reference gossip_sgd.py:654-666)."""
    if backend == "gloo":
        assert network_interface_type == "ethernet"
        os.environ["GLOO_SOCKET_IFNAME"] = get_tcp_interface_name(
            network_interface_type
        )
    elif network_interface_type == "ethernet":
        if backend == "nccl":
            os.environ["NCCL_SOCKET_IFNAME"] = get_tcp_interface_name(
                network_interface_type
            )
            os.environ["NCCL_IB_DISABLE"] = "1"
        else:
            raise NotImplementedError(backend)
