CREATE TABLE lg (svc STRING, ts TIMESTAMP TIME INDEX, msg STRING, lat DOUBLE, PRIMARY KEY (svc)) WITH ('append_mode'='true');
INSERT INTO lg (svc, ts, msg, lat) VALUES ('a', 1000, 'connection timeout error', 5.0), ('b', 2000, 'request ok', 1.0), ('a', 3000, 'disk error detected', 2.0);
SELECT count(*) FROM lg WHERE matches(msg, 'error');
SELECT svc, msg FROM lg WHERE matches(msg, 'timeout') ORDER BY ts;
SELECT count(*) FROM lg WHERE matches(msg, 'error') AND lat > 3;
ADMIN flush_table('lg');
SELECT count(*) FROM lg WHERE matches(msg, 'error');
