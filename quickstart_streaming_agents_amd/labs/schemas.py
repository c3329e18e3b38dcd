"""Topic value schemas for the four labs.

Field-for-field the schemas the reference publishes (compatibility
contract): lab1 scripts/publish_lab1_data.py:50-108, lab3
terraform/lab3-agentic-fleet-management/main.tf:301-312, lab4
scripts/lab4_datagen.py:100-123, docs scripts/publish_docs.py:63-109.
"""

NAMESPACE = "org.apache.flink.avro.generated.record"


def _record(name, fields):
    return {"type": "record", "name": name, "namespace": NAMESPACE, "fields": fields}


def _ts(name):
    return {"name": name, "type": {"type": "long", "logicalType": "timestamp-millis"}}


def _opt(name):
    return {"name": name, "type": ["null", "string"], "default": None}


CUSTOMERS = _record("customers_value", [
    {"name": "customer_id", "type": "string"},
    {"name": "customer_email", "type": "string"},
    {"name": "customer_name", "type": "string"},
    {"name": "state", "type": "string"},
    _ts("updated_at"),
])

PRODUCTS = _record("products_value", [
    {"name": "product_id", "type": "string"},
    {"name": "product_name", "type": "string"},
    {"name": "price", "type": "double"},
    {"name": "department", "type": "string"},
    _ts("updated_at"),
])

ORDERS = _record("orders_value", [
    {"name": "order_id", "type": "string"},
    {"name": "customer_id", "type": "string"},
    {"name": "product_id", "type": "string"},
    {"name": "price", "type": "double"},
    _ts("order_ts"),
])

RIDE_REQUESTS = _record("ride_requests_value", [
    {"name": "request_id", "type": "string"},
    {"name": "customer_email", "type": "string"},
    {"name": "pickup_zone", "type": "string"},
    {"name": "drop_off_zone", "type": "string"},
    {"name": "price", "type": "double"},
    {"name": "number_of_passengers", "type": "int"},
    _ts("request_ts"),
])

CLAIMS = _record("claims_value", [
    {"name": "claim_id", "type": "string"},
    _opt("applicant_name"),
    {"name": "city", "type": "string"},
    _opt("is_primary_residence"),
    _opt("damage_assessed"),
    {"name": "claim_amount", "type": "string"},
    _opt("has_insurance"),
    _opt("insurance_amount"),
    _opt("claim_narrative"),
    _opt("assessment_date"),
    _opt("disaster_date"),
    _opt("previous_claims_count"),
    _opt("last_claim_date"),
    _opt("assessment_source"),
    _opt("shared_account"),
    _opt("shared_phone"),
    _ts("claim_timestamp"),
])

QUERIES = _record("queries_value", [
    {"name": "query", "type": "string"},
])

DOCUMENTS = _record("documents_value", [
    {"name": "document_id", "type": "string"},
    {"name": "title", "type": ["null", "string"], "default": None},
    {"name": "chunk", "type": "string"},
    {"name": "pages", "type": ["null", "string"], "default": None},
    {"name": "section_reference", "type": ["null", "string"], "default": None},
    {"name": "fraud_categories", "type": ["null", {"type": "array", "items": "string"}], "default": None},
    {"name": "policy_keywords", "type": ["null", {"type": "array", "items": "string"}], "default": None},
    {"name": "char_count", "type": ["null", "int"], "default": None},
])

# New Orleans zones the lab3 datagen forks over
# (terraform/lab3-.../data-gen/zones/all-zones.json; surge zone = French Quarter)
LAB3_ZONES = [
    "French Quarter",
    "Marigny",
    "Bywater",
    "Warehouse District",
    "Uptown",
    "Garden District",
    "Central Business District (CBD)",
]
LAB3_SURGE_ZONE = "French Quarter"

# Florida cities for lab4; Naples spikes in the final 2 days
# (LAB4-Walkthrough.md:66-68)
LAB4_CITIES = [
    "Naples", "Fort Myers", "Cape Coral", "Sarasota",
    "Tampa", "Orlando", "Miami", "Jacksonville",
]
LAB4_SPIKE_CITY = "Naples"
