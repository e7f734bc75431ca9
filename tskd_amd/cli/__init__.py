"""Pipeline entrypoints with the reference's flags and wire contract.

  python -m tskd_amd.cli.sendstream    — replay producer  (bin/sendStream.py)
  python -m tskd_amd.cli.processstream — preprocessor      (bin/processStream.py)
  python -m tskd_amd.cli.predictstream — inference service (bin/predictStream.py)
  python -m tskd_amd.cli.plotdata      — dashboard         (bin/plotData.py)
  python -m tskd_amd.cli.makedata      — synthetic CSV gen (data/makeData.py)

Every stage is a separate OS process sharing the bus directory (the
docker-compose topology without docker); `--speed` compresses every
window/trigger/sleep duration exactly like the reference.
"""
