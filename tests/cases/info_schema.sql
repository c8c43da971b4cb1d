CREATE TABLE isc (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
SELECT table_name FROM information_schema.tables WHERE table_name = 'isc';
SELECT column_name, semantic_type FROM information_schema.columns WHERE table_name = 'isc';
SHOW DATABASES
