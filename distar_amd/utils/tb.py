"""TensorBoard-compatible event-file writer (no tensorboardX dependency).

The reference logs through tensorboardX (`ctools/utils/log_helper.py:23-64`);
this image has no tensorboard package, so this module writes the format
directly: TFRecord framing (length + masked CRC32C) around `Event` protobuf
messages built with google.protobuf dynamic descriptors (field numbers from
tensorboard's event.proto / summary.proto).  Scalars only — the drop-in
surface the stack uses (`add_scalar`), readable by standard TensorBoard.
"""
import os
import socket
import struct
import threading
import time

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

# ------------------------------------------------------------------ crc32c
_CRC_TABLE = []


def _build_crc_table():
    poly = 0x82F63B78                      # Castagnoli, reflected
    for i in range(256):
        crc = i
        for _ in range(8):
            crc = (crc >> 1) ^ poly if crc & 1 else crc >> 1
        _CRC_TABLE.append(crc)


_build_crc_table()


def crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ------------------------------------------------------------- event proto
_F = descriptor_pb2.FieldDescriptorProto
_LOCK = threading.Lock()
_CLASSES = None


def _classes():
    global _CLASSES
    with _LOCK:
        if _CLASSES is not None:
            return _CLASSES
        fdp = descriptor_pb2.FileDescriptorProto()
        fdp.name = 'distar_tb.proto'
        fdp.package = 'DistarTB'
        fdp.syntax = 'proto2'
        v = fdp.message_type.add()
        v.name = 'SummaryValue'
        f = v.field.add()
        f.name, f.number, f.type, f.label = 'tag', 1, _F.TYPE_STRING, \
            _F.LABEL_OPTIONAL
        f = v.field.add()
        f.name, f.number, f.type, f.label = 'simple_value', 2, \
            _F.TYPE_FLOAT, _F.LABEL_OPTIONAL
        s = fdp.message_type.add()
        s.name = 'Summary'
        f = s.field.add()
        f.name, f.number, f.type, f.label = 'value', 1, _F.TYPE_MESSAGE, \
            _F.LABEL_REPEATED
        f.type_name = '.DistarTB.SummaryValue'
        e = fdp.message_type.add()
        e.name = 'Event'
        for name, num, typ in (('wall_time', 1, _F.TYPE_DOUBLE),
                               ('step', 2, _F.TYPE_INT64),
                               ('file_version', 3, _F.TYPE_STRING)):
            f = e.field.add()
            f.name, f.number, f.type, f.label = name, num, typ, \
                _F.LABEL_OPTIONAL
        f = e.field.add()
        f.name, f.number, f.type, f.label = 'summary', 5, _F.TYPE_MESSAGE, \
            _F.LABEL_OPTIONAL
        f.type_name = '.DistarTB.Summary'
        pool = descriptor_pool.DescriptorPool()
        pool.Add(fdp)
        try:
            classes = message_factory.GetMessageClassesForFiles(
                ['distar_tb.proto'], pool)
        except AttributeError:             # protobuf < 4.22
            classes = message_factory.MessageFactory(pool).GetMessages(
                ['distar_tb.proto'])
        _CLASSES = {k.split('.')[-1]: v for k, v in classes.items()}
        return _CLASSES


class SummaryWriter:
    """tensorboardX-shaped scalar writer (`add_scalar`, `flush`, `close`)."""

    def __init__(self, log_dir):
        os.makedirs(log_dir, exist_ok=True)
        fname = 'events.out.tfevents.{:d}.{}'.format(
            int(time.time()), socket.gethostname())
        self._path = os.path.join(log_dir, fname)
        self._f = open(self._path, 'wb')
        self._lock = threading.Lock()
        ev = _classes()['Event']()
        ev.wall_time = time.time()
        ev.file_version = 'brain.Event:2'
        self._write_record(ev.SerializeToString())

    def _write_record(self, data: bytes):
        header = struct.pack('<Q', len(data))
        self._f.write(header)
        self._f.write(struct.pack('<I', _masked_crc(header)))
        self._f.write(data)
        self._f.write(struct.pack('<I', _masked_crc(data)))

    def add_scalar(self, tag, value, global_step=0, walltime=None):
        cls = _classes()
        ev = cls['Event']()
        ev.wall_time = walltime if walltime is not None else time.time()
        ev.step = int(global_step)
        val = ev.summary.value.add()
        val.tag = str(tag)
        val.simple_value = float(value)
        with self._lock:
            self._write_record(ev.SerializeToString())

    def add_scalars(self, prefix, tag_value_dict, global_step=0):
        for tag, value in tag_value_dict.items():
            self.add_scalar(f'{prefix}/{tag}', value, global_step)

    def flush(self):
        with self._lock:
            self._f.flush()

    def close(self):
        with self._lock:
            self._f.flush()
            self._f.close()
