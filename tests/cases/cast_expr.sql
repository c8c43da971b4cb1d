SELECT 1 + 2 * 3 AS arith;
SELECT 10 / 4 AS division, 10 % 3 AS modulo;
SELECT -5 AS neg, abs(-5) AS absval;
SELECT round(3.14159, 2) AS r2, floor(2.7) AS f, ceil(2.1) AS c;
SELECT sqrt(16.0) AS s, pow(2.0, 10.0) AS p;
SELECT CASE WHEN 1 < 2 THEN 'yes' ELSE 'no' END AS cmp
