#!/usr/bin/env python3
"""Report which scheduler/worker/job ports are in use on this node
(reference scripts/utils port monitor): the control-plane ports (50070,
50061+) and the distributed-job rendezvous range (60570+)."""

import argparse
import socket


def port_open(port, host="127.0.0.1"):
    with socket.socket() as s:
        s.settimeout(0.2)
        return s.connect_ex((host, port)) == 0


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--job_ports", type=int, default=32,
                   help="how many job rendezvous ports to probe from 60570")
    args = p.parse_args()
    print(f"scheduler 50070: {'OPEN' if port_open(50070) else 'free'}")
    for port in range(50061, 50069):
        if port_open(port):
            print(f"worker {port}: OPEN")
    busy = [p_ for p_ in range(60570, 60570 + args.job_ports) if port_open(p_)]
    print(f"job ports busy ({len(busy)}): {busy}")


if __name__ == "__main__":
    main()
