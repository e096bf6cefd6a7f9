"""Command-line interface: `python -m spark_tfrecord_amd <cmd> ...`.

    count    PATH                    record count (frame scan only)
    schema   PATH [--record-type T]  inferred schema
    validate PATH                    CRC-check every frame of every file
    head     PATH [-n N]             first N records as JSON lines
    convert  SRC DST [--codec C]     re-encode a dataset (e.g. gzip <-> plain)
"""

import argparse
import json
import sys


def main(argv=None):
    ap = argparse.ArgumentParser(prog="spark_tfrecord_amd", description=__doc__)
    ap.add_argument("--engine", default="auto", choices=["auto", "cpu", "gpu"])
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("count", help="record count without decoding")
    p.add_argument("path")

    p = sub.add_parser("schema", help="inferred schema")
    p.add_argument("path")
    p.add_argument("--record-type", default="Example",
                   choices=["Example", "SequenceExample", "ByteArray"])

    p = sub.add_parser("validate", help="CRC-check every frame")
    p.add_argument("path")

    p = sub.add_parser("head", help="first N records as JSON lines")
    p.add_argument("path")
    p.add_argument("-n", type=int, default=10)
    p.add_argument("--record-type", default="Example",
                   choices=["Example", "SequenceExample", "ByteArray"])

    p = sub.add_parser("convert", help="re-encode a dataset")
    p.add_argument("src")
    p.add_argument("dst")
    p.add_argument("--codec", default=None)
    p.add_argument("--record-type", default="Example",
                   choices=["Example", "SequenceExample", "ByteArray"])
    p.add_argument("--num-shards", type=int, default=1)
    p.add_argument("--mode", default="errorifexists")

    args = ap.parse_args(argv)
    import spark_tfrecord_amd as stf

    if args.cmd == "count":
        print(stf.count_tfrecord(args.path, engine=args.engine))
    elif args.cmd == "schema":
        from .io.reader import infer_schema_of_paths
        from .io import paths as P

        files = P.list_data_files(args.path)
        s = infer_schema_of_paths(files, args.record_type,
                                  stf.engine.resolve_engine(args.engine))
        print(s.simple_string())
    elif args.cmd == "validate":
        rep = stf.validate_tfrecord(args.path, engine=args.engine)
        print(json.dumps({
            "ok": rep.ok, "files": len(rep.files), "records": rep.records,
            "bad": [{"path": f.path, "error": f.error}
                    for f in rep.files if not f.ok],
        }, indent=2))
        return 0 if rep.ok else 1
    elif args.cmd == "head":
        df = stf.read_tfrecord(args.path, record_type=args.record_type,
                               engine=args.engine)
        for row in df.collect()[: args.n]:
            print(json.dumps({k: (v if not isinstance(v, bytes) else v.hex())
                              for k, v in row.items()}, default=str))
    elif args.cmd == "convert":
        df = stf.read_tfrecord(args.src, record_type=args.record_type,
                               engine=args.engine)
        stf.write_tfrecord(df, args.dst, record_type=args.record_type,
                           codec=args.codec, mode=args.mode,
                           num_shards=args.num_shards, engine=args.engine)
        print(f"wrote {df.count()} records to {args.dst}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
