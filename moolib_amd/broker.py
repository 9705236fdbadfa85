"""Standalone broker: ``python -m moolib_amd.broker [addr:port]``.

Capability parity with the reference's py/moolib/broker.py.
"""
import argparse
import time

import moolib_amd

DEFAULT_PORT = 4431

parser = argparse.ArgumentParser(description="Run a moolib_amd broker")
parser.add_argument(
    "address",
    nargs="?",
    default="0.0.0.0:%i" % DEFAULT_PORT,
    type=str,
    metavar="addr:port",
    help="Broker server address to listen on.",
)


def main():
    flags = parser.parse_args()

    broker_rpc = moolib_amd.Rpc()
    broker_rpc.set_name("broker")
    broker = moolib_amd.Broker(broker_rpc)
    bound = broker_rpc.listen(flags.address)

    print("Broker listening at %s" % ", ".join(bound), flush=True)

    try:
        while True:
            broker.update()
            time.sleep(0.25)
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
