"""Starts a standalone Vizier gRPC server.

Parity with the reference's demos/run_vizier_server.py (argparse instead
of absl flags).

Usage:
  python demos/run_vizier_server.py --host localhost --port 28080 \
      --database_url sqlite:///vizier.db
"""

import argparse
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

from vizier_amd.service import DefaultVizierServer


def main() -> None:
  parser = argparse.ArgumentParser(description=__doc__)
  parser.add_argument('--host', default='localhost')
  parser.add_argument('--port', type=int, default=None,
                      help='Port (default: pick an unused one).')
  parser.add_argument('--database_url', default='sqlite:///:memory:',
                      help='SQLAlchemy URL, or "ram" for the in-RAM '
                           'datastore.')
  args = parser.parse_args()
  logging.basicConfig(level=logging.INFO)

  database_url = None if args.database_url == 'ram' else args.database_url
  server = DefaultVizierServer(host=args.host, port=args.port,
                               database_url=database_url)
  print(f'Vizier server listening at {server.endpoint}')
  try:
    while True:
      time.sleep(3600)
  except KeyboardInterrupt:
    server.stop(grace=1.0)


if __name__ == '__main__':
  main()
