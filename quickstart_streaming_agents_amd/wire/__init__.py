from .avro import Schema, serialize, deserialize, peek_schema_id
from .registry import SchemaRegistry
from .topics import Broker, Topic, Record, AvroProducer, AvroConsumer

__all__ = [
    "Schema", "serialize", "deserialize", "peek_schema_id",
    "SchemaRegistry", "Broker", "Topic", "Record", "AvroProducer", "AvroConsumer",
]
